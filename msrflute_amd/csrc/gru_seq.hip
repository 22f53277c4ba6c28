// Fused GRU sequence recurrence for gfx950 (nlg_gru task).
//
// Same design as lstm_seq.hip: one 512-thread workgroup per batch row
// runs the whole T-step recurrence; h_{t-1} is wave-broadcast from LDS;
// the forward reads W_hh transposed for coalesced lanes.  Cell math
// follows the reference GRU2 (experiments/nlg_gru/model.py:20-28):
//   gh = W_hh h + b_hh            (three blocks r, i/z, n)
//   r = sig(gi_r + gh_r); z = sig(gi_z + gh_z)
//   n = tanh(gi_n + r*gh_n);  h' = n + z*(h - n)
// gi = W_ih x + b_ih is precomputed for ALL steps by one caller GEMM.
//
// Saved for backward: r, z, n (activated, [B,T,3H]) and gh_n [B,T,H].

#include <hip/hip_runtime.h>

#define GRU_H 512

__device__ inline float gsig(float x) { return 1.f / (1.f + __expf(-x)); }

// gi:     [B, T, 3H]
// whh_t:  [H, 3H]   (transposed W_hh)
// b_hh:   [3H]
// h_seq:  [B, T, H] out
// gates:  [B, T, 3H] out (activated r, z, n)
// ghn:    [B, T, H] out (pre-activation hidden contribution of the n gate)
__global__ __launch_bounds__(GRU_H)
void k_gru_seq_fwd(const float* __restrict__ gi,
                   const float* __restrict__ whh_t,
                   const float* __restrict__ b_hh,
                   float* __restrict__ h_seq,
                   float* __restrict__ gates,
                   float* __restrict__ ghn, int B, int T) {
  __shared__ float h_prev[GRU_H];
  int h = threadIdx.x;
  int b = blockIdx.x;
  h_prev[h] = 0.f;
  // float4-packed weights ([H/4, 3H, 4]: element (kk, j, d) =
  // W_hh[j, 4kk+d]) cut the latency-bound per-step loads 4x, and the
  // first GWREG groups of each gate column stay register-resident
  // across ALL T steps (W_hh is time-invariant) — the lstm_seq.hip
  // recipe, measured +34% there
  const float4* wp = reinterpret_cast<const float4*>(whh_t);
  const float4* hp4 = reinterpret_cast<const float4*>(h_prev);
  #define GWREG 8
  float4 wrr[GWREG], wzr[GWREG], wnr[GWREG];
  #pragma unroll
  for (int kk = 0; kk < GWREG; ++kk) {
    wrr[kk] = wp[(long long)kk * 3 * GRU_H + h];
    wzr[kk] = wp[(long long)kk * 3 * GRU_H + GRU_H + h];
    wnr[kk] = wp[(long long)kk * 3 * GRU_H + 2 * GRU_H + h];
  }
  __syncthreads();
  float br = b_hh[h], bz = b_hh[GRU_H + h], bn = b_hh[2 * GRU_H + h];
  for (int t = 0; t < T; ++t) {
    const float* gr = gi + ((long long)b * T + t) * 3 * GRU_H;
    float sr = br, sz = bz, sn = bn;
    #pragma unroll
    for (int kk = 0; kk < GWREG; ++kk) {
      float4 hv = hp4[kk];
      sr = fmaf(wrr[kk].x, hv.x, fmaf(wrr[kk].y, hv.y,
           fmaf(wrr[kk].z, hv.z, fmaf(wrr[kk].w, hv.w, sr))));
      sz = fmaf(wzr[kk].x, hv.x, fmaf(wzr[kk].y, hv.y,
           fmaf(wzr[kk].z, hv.z, fmaf(wzr[kk].w, hv.w, sz))));
      sn = fmaf(wnr[kk].x, hv.x, fmaf(wnr[kk].y, hv.y,
           fmaf(wnr[kk].z, hv.z, fmaf(wnr[kk].w, hv.w, sn))));
    }
    #pragma unroll 8
    for (int kk = GWREG; kk < GRU_H / 4; ++kk) {
      float4 hv = hp4[kk];
      float4 w0 = wp[(long long)kk * 3 * GRU_H + h];
      float4 w1 = wp[(long long)kk * 3 * GRU_H + GRU_H + h];
      float4 w2 = wp[(long long)kk * 3 * GRU_H + 2 * GRU_H + h];
      sr = fmaf(w0.x, hv.x, fmaf(w0.y, hv.y,
           fmaf(w0.z, hv.z, fmaf(w0.w, hv.w, sr))));
      sz = fmaf(w1.x, hv.x, fmaf(w1.y, hv.y,
           fmaf(w1.z, hv.z, fmaf(w1.w, hv.w, sz))));
      sn = fmaf(w2.x, hv.x, fmaf(w2.y, hv.y,
           fmaf(w2.z, hv.z, fmaf(w2.w, hv.w, sn))));
    }
    float hp = h_prev[h];
    float r = gsig(gr[h] + sr);
    float z = gsig(gr[GRU_H + h] + sz);
    float n = tanhf(gr[2 * GRU_H + h] + r * sn);
    float hn = n + z * (hp - n);
    long long base = ((long long)b * T + t) * 3 * GRU_H;
    gates[base + h] = r;
    gates[base + GRU_H + h] = z;
    gates[base + 2 * GRU_H + h] = n;
    ghn[((long long)b * T + t) * GRU_H + h] = sn;
    h_seq[((long long)b * T + t) * GRU_H + h] = hn;
    __syncthreads();
    h_prev[h] = hn;
    __syncthreads();
  }
}

// Backward: produces dgh [B,T,3H] (grads wrt the PRE-activation hidden
// contributions — caller turns them into dW_hh/db_hh/dh via GEMMs) and
// dgi [B,T,3H] (grads of the input projection).
__global__ void k_gru_seq_bwd(const float* __restrict__ gates,
                              const float* __restrict__ ghn,
                              const float* __restrict__ h_seq,
                              const float* __restrict__ w_hh,
                              const float* __restrict__ dh_out,
                              float* __restrict__ dgi,
                              float* __restrict__ dgh, int B, int T) {
  __shared__ float dgh_l[3 * GRU_H];
  int h = threadIdx.x;
  int b = blockIdx.x;
  float dh_rec = 0.f;
  for (int t = T - 1; t >= 0; --t) {
    long long base = ((long long)b * T + t) * 3 * GRU_H;
    long long hbase = ((long long)b * T + t) * GRU_H;
    float r = gates[base + h];
    float z = gates[base + GRU_H + h];
    float n = gates[base + 2 * GRU_H + h];
    float sn = ghn[hbase + h];
    float hp = (t > 0) ? h_seq[hbase - GRU_H + h] : 0.f;
    float dh = dh_out[hbase + h] + dh_rec;
    float dn = dh * (1.f - z);
    float dz = dh * (hp - n);
    float dhp_direct = dh * z;
    float dan = dn * (1.f - n * n);        // pre-tanh n grad
    float dr = dan * sn;
    float dghn = dan * r;
    float daz = dz * z * (1.f - z);
    float dar = dr * r * (1.f - r);
    // input-projection grads (gate order r, z, n)
    dgi[base + h] = dar;
    dgi[base + GRU_H + h] = daz;
    dgi[base + 2 * GRU_H + h] = dan;
    dgh[base + h] = dar;
    dgh[base + GRU_H + h] = daz;
    dgh[base + 2 * GRU_H + h] = dghn;
    __syncthreads();
    dgh_l[h] = dar;
    dgh_l[GRU_H + h] = daz;
    dgh_l[2 * GRU_H + h] = dghn;
    __syncthreads();
    {
      const float4* wpB = reinterpret_cast<const float4*>(w_hh);
      const float4* dg4 = reinterpret_cast<const float4*>(dgh_l);
      float s0 = dhp_direct, s1 = 0.f, s2 = 0.f, s3 = 0.f;
      #pragma unroll 8
      for (int jg = 0; jg < 3 * GRU_H / 4; ++jg) {
        float4 wv = wpB[(long long)jg * GRU_H + h];
        float4 dv = dg4[jg];
        s0 = fmaf(wv.x, dv.x, s0);
        s1 = fmaf(wv.y, dv.y, s1);
        s2 = fmaf(wv.z, dv.z, s2);
        s3 = fmaf(wv.w, dv.w, s3);
      }
      dh_rec = (s0 + s1) + (s2 + s3);
    }
    __syncthreads();
  }
}

extern "C" {

void launch_gru_seq_fwd(const float* gi, const float* whh_t,
                        const float* b_hh, float* h_seq, float* gates,
                        float* ghn, int B, int T, hipStream_t s) {
  hipLaunchKernelGGL(k_gru_seq_fwd, dim3(B), dim3(GRU_H), 0, s,
                     gi, whh_t, b_hh, h_seq, gates, ghn, B, T);
}

void launch_gru_seq_bwd(const float* gates, const float* ghn,
                        const float* h_seq, const float* w_hh,
                        const float* dh_out, float* dgi, float* dgh,
                        int B, int T, hipStream_t s) {
  hipLaunchKernelGGL(k_gru_seq_bwd, dim3(B), dim3(GRU_H), 0, s,
                     gates, ghn, h_seq, w_hh, dh_out, dgi, dgh, B, T);
}

}  // extern "C"
