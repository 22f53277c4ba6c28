// Fused LSTM sequence recurrence for gfx950 (nlp_rnn_fedshakespeare /
// BASELINE benchmark task 4).
//
// MIOpen's RNN path is both slow at FL batch sizes (~9 ms per batch-4
// step sequence) and segfaults under hipGraph capture, so the eager
// Shakespeare client ran at reference speed.  These two kernels run the
// ENTIRE T-step recurrence of one layer in one launch each:
//
//   fwd: given X_proj = x@W_ih^T + b_ih + b_hh (one hipBLASLt GEMM for
//        all steps, done by the caller), loop t: gates = X_proj[t] +
//        W_hh @ h_{t-1}; i,f,g,o activations; c_t, h_t.  Saves activated
//        gates + c for backward.
//   bwd: reverse BPTT loop producing the PRE-activation gate grads
//        (= grads of X_proj, from which the caller gets dW_ih, db and dx
//        with GEMMs) and dh via W_hh^T inside the loop.
//
// Parallel shape: the recurrence is serial in t but independent across
// the batch — each 1024-thread workgroup owns ROWS_PER_WG batch rows
// (thread = (b_local, h)); h_{t-1} lives in LDS and is wave-broadcast in
// the k-loop.  Weights stream from L2 (W_hh for H=256 is 1 MB).
// Gate order follows torch: i, f, g, o.

#include <hip/hip_runtime.h>

#define LSTM_H 256           // hidden size of the benchmark model

__device__ inline float sigf(float x) { return 1.f / (1.f + __expf(-x)); }

// One 256-thread workgroup per batch row (thread = hidden unit): batch
// rows run on separate CUs, h_{t-1} is wave-broadcast from LDS, and the
// forward reads W_hh TRANSPOSED ([H, 4H], passed by the caller) so lanes
// with consecutive h hit consecutive addresses.
//
// xp:      [B, T, 4H]  (pre-activation input projection, gate order ifgo)
// w_hh_t:  [H, 4H]     (transposed W_hh)
// h_seq:   [B, T, H]   out
// gates:   [B, T, 4H]  out (ACTIVATED i,f,g,o — saved for backward)
// c_seq:   [B, T, H]   out (cell states)
// 1024 threads per row (16 waves on one CU): thread = gate column
// j = g*H + h, so the per-step GEMV reads w_hh_t[k*4H + j] fully
// coalesced with 4 waves per SIMD hiding the L2 latency (the 256-thread
// form was latency-bound at ~1 ms per T=80 call; this is ~14x less
// exposed latency per step).  Unroll 8 batches the loads.
__global__ __launch_bounds__(1024)
void k_lstm_seq_fwd(const float* __restrict__ xp,
                    const float* __restrict__ w_hh_t,
                    float* __restrict__ h_seq,
                    float* __restrict__ gates,
                    float* __restrict__ c_seq, int B, int T) {
  __shared__ float h_prev[LSTM_H];
  __shared__ float act_l[4 * LSTM_H];
  int tid = threadIdx.x;           // gate column j = g*H + h
  int g = tid >> 8;
  int b = blockIdx.x;
  const float4* wp = reinterpret_cast<const float4*>(w_hh_t);
  const float4* hp4 = reinterpret_cast<const float4*>(h_prev);
  if (tid < LSTM_H) h_prev[tid] = 0.f;
  float c = 0.f;                   // live in threads tid < H only
  // first 16 float4 weight groups register-resident across all T steps
  // (W_hh is time-invariant; same trick as the batched variant below —
  // 16 is the 128-VGPR/16-wave occupancy limit, 24 spills)
  float4 wregS[16];
  #pragma unroll
  for (int kk = 0; kk < 16; ++kk)
    wregS[kk] = wp[(long long)kk * 4 * LSTM_H + tid];
  __syncthreads();
  for (int t = 0; t < T; ++t) {
    const float* xr = xp + ((long long)b * T + t) * 4 * LSTM_H;
    // float4-packed weights ([H/4, 4H, 4]: one b128 load covers 4 k's)
    // + 4 independent accumulator chains: the b32 form was bound by
    // exposed L2 latency x load count (256 loads/step/thread -> 64)
    float s0 = xr[tid], s1 = 0.f, s2 = 0.f, s3 = 0.f;
    #pragma unroll
    for (int kk = 0; kk < 16; ++kk) {
      float4 hv = hp4[kk];
      s0 = fmaf(wregS[kk].x, hv.x, s0);
      s1 = fmaf(wregS[kk].y, hv.y, s1);
      s2 = fmaf(wregS[kk].z, hv.z, s2);
      s3 = fmaf(wregS[kk].w, hv.w, s3);
    }
    #pragma unroll 16
    for (int kk = 16; kk < LSTM_H / 4; ++kk) {
      float4 wv = wp[(long long)kk * 4 * LSTM_H + tid];
      float4 hv = hp4[kk];
      s0 = fmaf(wv.x, hv.x, s0);
      s1 = fmaf(wv.y, hv.y, s1);
      s2 = fmaf(wv.z, hv.z, s2);
      s3 = fmaf(wv.w, hv.w, s3);
    }
    float s = (s0 + s1) + (s2 + s3);
    float act = (g == 2) ? tanhf(s) : sigf(s);
    long long base = ((long long)b * T + t) * 4 * LSTM_H;
    gates[base + tid] = act;       // ACTIVATED i,f,g,o (order ifgo)
    act_l[tid] = act;
    __syncthreads();
    if (tid < LSTM_H) {
      float i = act_l[tid], f = act_l[LSTM_H + tid];
      float gg = act_l[2 * LSTM_H + tid], o = act_l[3 * LSTM_H + tid];
      c = f * c + i * gg;
      float hn = o * tanhf(c);
      c_seq[((long long)b * T + t) * LSTM_H + tid] = c;
      h_seq[((long long)b * T + t) * LSTM_H + tid] = hn;
      h_prev[tid] = hn;
    }
    __syncthreads();
  }
}

// Backward: dh_out [B,T,H] upstream grads of h_seq; produces
// dg_pre [B,T,4H] (pre-activation gate grads == grads of xp).
// W_hh in its ORIGINAL [4H, H] layout: w_hh[j*H + h] is coalesced across
// lanes (consecutive h).
__global__ __launch_bounds__(1024)
void k_lstm_seq_bwd(const float* __restrict__ gates,
                    const float* __restrict__ c_seq,
                    const float* __restrict__ w_hh,
                    const float* __restrict__ dh_out,
                    float* __restrict__ dg_pre, int B, int T) {
  __shared__ float dg_l[4 * LSTM_H];
  __shared__ float part[4 * LSTM_H];
  __shared__ float dh_rec_l[LSTM_H];
  int tid = threadIdx.x;
  int h = tid & (LSTM_H - 1), q = tid >> 8;  // partial-dot quarter
  int b = blockIdx.x;
  const float4* wpB = reinterpret_cast<const float4*>(w_hh);
  const float4* dgp4 = reinterpret_cast<const float4*>(dg_l);
  float dc = 0.f;                            // live in threads tid < H
  if (tid < LSTM_H) dh_rec_l[tid] = 0.f;
  // register-resident first 16 weight groups of this thread's quarter
  float4 wregSB[16];
  #pragma unroll
  for (int jj = 0; jj < 16; ++jj)
    wregSB[jj] = wpB[(long long)(q * (LSTM_H / 4) + jj) * LSTM_H + h];
  __syncthreads();
  for (int t = T - 1; t >= 0; --t) {
    long long base = ((long long)b * T + t) * 4 * LSTM_H;
    long long cbase = ((long long)b * T + t) * LSTM_H;
    if (tid < LSTM_H) {
      float i = gates[base + tid];
      float f = gates[base + LSTM_H + tid];
      float g = gates[base + 2 * LSTM_H + tid];
      float o = gates[base + 3 * LSTM_H + tid];
      float ct = c_seq[cbase + tid];
      float cprev = (t > 0) ? c_seq[cbase - LSTM_H + tid] : 0.f;
      float tc = tanhf(ct);
      float dh = dh_out[cbase + tid] + dh_rec_l[tid];
      float do_ = dh * tc * o * (1.f - o);
      dc = dc + dh * o * (1.f - tc * tc);
      float di = dc * g * i * (1.f - i);
      float df = dc * cprev * f * (1.f - f);
      float dg = dc * i * (1.f - g * g);
      dc = dc * f;
      dg_l[tid] = di;
      dg_l[LSTM_H + tid] = df;
      dg_l[2 * LSTM_H + tid] = dg;
      dg_l[3 * LSTM_H + tid] = do_;
      dg_pre[base + tid] = di;
      dg_pre[base + LSTM_H + tid] = df;
      dg_pre[base + 2 * LSTM_H + tid] = dg;
      dg_pre[base + 3 * LSTM_H + tid] = do_;
    }
    __syncthreads();
    // dh_rec = W_hh^T dg: thread (q, h) sums its 256-row quarter with
    // lane-coalesced w_hh[j*H + h] reads; 16 waves hide the L2 latency
    float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
    #pragma unroll
    for (int jj = 0; jj < 16; ++jj) {
      float4 dv = dgp4[q * (LSTM_H / 4) + jj];
      s0 = fmaf(wregSB[jj].x, dv.x, s0);
      s1 = fmaf(wregSB[jj].y, dv.y, s1);
      s2 = fmaf(wregSB[jj].z, dv.z, s2);
      s3 = fmaf(wregSB[jj].w, dv.w, s3);
    }
    #pragma unroll 16
    for (int jj = 16; jj < LSTM_H / 4; ++jj) {
      int jg = q * (LSTM_H / 4) + jj;    // float4 group of 4 j's
      float4 wv = wpB[(long long)jg * LSTM_H + h];
      float4 dv = dgp4[jg];
      s0 = fmaf(wv.x, dv.x, s0);
      s1 = fmaf(wv.y, dv.y, s1);
      s2 = fmaf(wv.z, dv.z, s2);
      s3 = fmaf(wv.w, dv.w, s3);
    }
    float s = (s0 + s1) + (s2 + s3);
    part[tid] = s;
    __syncthreads();
    if (tid < LSTM_H)
      dh_rec_l[tid] = part[tid] + part[LSTM_H + tid]
                      + part[2 * LSTM_H + tid] + part[3 * LSTM_H + tid];
    __syncthreads();
  }
}

extern "C" {

void launch_lstm_seq_fwd(const float* xp, const float* w_hh_t, float* h_seq,
                         float* gates, float* c_seq, int B, int T, int H,
                         hipStream_t s) {
  hipLaunchKernelGGL(k_lstm_seq_fwd, dim3(B), dim3(4 * LSTM_H), 0, s,
                     xp, w_hh_t, h_seq, gates, c_seq, B, T);
}

void launch_lstm_seq_bwd(const float* gates, const float* c_seq,
                         const float* w_hh, const float* dh_out,
                         float* dg_pre, int B, int T, int H, hipStream_t s) {
  hipLaunchKernelGGL(k_lstm_seq_bwd, dim3(B), dim3(4 * LSTM_H), 0, s,
                     gates, c_seq, w_hh, dh_out, dg_pre, B, T);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// cross-client batched variants (Shakespeare mega round): the grid is
// ALL clients' rows (row r belongs to client r / rows_per_client) and
// each block indexes its client's weights in the K-stacked buffers —
// one launch covers the whole cohort's recurrence instead of one
// kernel per client on co-scheduling-limited streams.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(1024)
void k_lstm_seq_fwd_b(const float* __restrict__ xp,
                      const float* __restrict__ w_hh_t_stack,
                      float* __restrict__ h_seq,
                      float* __restrict__ gates,
                      float* __restrict__ c_seq, int rows_per_client,
                      int T) {
  __shared__ float h_prev[LSTM_H];
  __shared__ float act_l[4 * LSTM_H];
  int tid = threadIdx.x;
  int g = tid >> 8;
  int b = blockIdx.x;
  const float* w_hh_t = w_hh_t_stack
      + (long long)(b / rows_per_client) * LSTM_H * 4 * LSTM_H;
  const float4* wp = reinterpret_cast<const float4*>(w_hh_t);
  const float4* hp4 = reinterpret_cast<const float4*>(h_prev);
  if (tid < LSTM_H) h_prev[tid] = 0.f;
  float c = 0.f;
  // weights are time-invariant: keep the first WREG float4 groups of
  // this thread's column resident in registers across ALL T steps
  // (removes 1/4 of the latency-bound per-step loads)
  #define WREG_F 16
  float4 wreg[WREG_F];
  #pragma unroll
  for (int kk = 0; kk < WREG_F; ++kk)
    wreg[kk] = wp[(long long)kk * 4 * LSTM_H + tid];
  __syncthreads();
  for (int t = 0; t < T; ++t) {
    const float* xr = xp + ((long long)b * T + t) * 4 * LSTM_H;
    // float4-packed weights ([H/4, 4H, 4]: one b128 load covers 4 k's)
    // + 4 independent accumulator chains: the b32 form was bound by
    // exposed L2 latency x load count (256 loads/step/thread -> 64)
    float s0 = xr[tid], s1 = 0.f, s2 = 0.f, s3 = 0.f;
    #pragma unroll
    for (int kk = 0; kk < WREG_F; ++kk) {
      float4 hv = hp4[kk];
      s0 = fmaf(wreg[kk].x, hv.x, s0);
      s1 = fmaf(wreg[kk].y, hv.y, s1);
      s2 = fmaf(wreg[kk].z, hv.z, s2);
      s3 = fmaf(wreg[kk].w, hv.w, s3);
    }
    #pragma unroll 16
    for (int kk = WREG_F; kk < LSTM_H / 4; ++kk) {
      float4 wv = wp[(long long)kk * 4 * LSTM_H + tid];
      float4 hv = hp4[kk];
      s0 = fmaf(wv.x, hv.x, s0);
      s1 = fmaf(wv.y, hv.y, s1);
      s2 = fmaf(wv.z, hv.z, s2);
      s3 = fmaf(wv.w, hv.w, s3);
    }
    float s = (s0 + s1) + (s2 + s3);
    float act = (g == 2) ? tanhf(s) : sigf(s);
    long long base = ((long long)b * T + t) * 4 * LSTM_H;
    gates[base + tid] = act;
    act_l[tid] = act;
    __syncthreads();
    if (tid < LSTM_H) {
      float i = act_l[tid], f = act_l[LSTM_H + tid];
      float gg = act_l[2 * LSTM_H + tid], o = act_l[3 * LSTM_H + tid];
      c = f * c + i * gg;
      float hn = o * tanhf(c);
      c_seq[((long long)b * T + t) * LSTM_H + tid] = c;
      h_seq[((long long)b * T + t) * LSTM_H + tid] = hn;
      h_prev[tid] = hn;
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(1024)
void k_lstm_seq_bwd_b(const float* __restrict__ gates,
                      const float* __restrict__ c_seq,
                      const float* __restrict__ w_hh_stack,
                      const float* __restrict__ dh_out,
                      float* __restrict__ dg_pre, int rows_per_client,
                      int T) {
  __shared__ float dg_l[4 * LSTM_H];
  __shared__ float part[4 * LSTM_H];
  __shared__ float dh_rec_l[LSTM_H];
  int tid = threadIdx.x;
  int h = tid & (LSTM_H - 1), q = tid >> 8;
  int b = blockIdx.x;
  const float* w_hh = w_hh_stack
      + (long long)(b / rows_per_client) * 4 * LSTM_H * LSTM_H;
  const float4* wpB = reinterpret_cast<const float4*>(w_hh);
  const float4* dgp4 = reinterpret_cast<const float4*>(dg_l);
  float dc = 0.f;
  if (tid < LSTM_H) dh_rec_l[tid] = 0.f;
  // register-resident first WREG_B weight groups of this thread's
  // quarter (time-invariant across the BPTT loop)
  #define WREG_B 16
  float4 wregB[WREG_B];
  #pragma unroll
  for (int jj = 0; jj < WREG_B; ++jj)
    wregB[jj] = wpB[(long long)(q * (LSTM_H / 4) + jj) * LSTM_H + h];
  __syncthreads();
  for (int t = T - 1; t >= 0; --t) {
    long long base = ((long long)b * T + t) * 4 * LSTM_H;
    long long cbase = ((long long)b * T + t) * LSTM_H;
    if (tid < LSTM_H) {
      float i = gates[base + tid];
      float f = gates[base + LSTM_H + tid];
      float g = gates[base + 2 * LSTM_H + tid];
      float o = gates[base + 3 * LSTM_H + tid];
      float ct = c_seq[cbase + tid];
      float cprev = (t > 0) ? c_seq[cbase - LSTM_H + tid] : 0.f;
      float tc = tanhf(ct);
      float dh = dh_out[cbase + tid] + dh_rec_l[tid];
      float do_ = dh * tc * o * (1.f - o);
      dc = dc + dh * o * (1.f - tc * tc);
      float di = dc * g * i * (1.f - i);
      float df = dc * cprev * f * (1.f - f);
      float dg = dc * i * (1.f - g * g);
      dc = dc * f;
      dg_l[tid] = di;
      dg_l[LSTM_H + tid] = df;
      dg_l[2 * LSTM_H + tid] = dg;
      dg_l[3 * LSTM_H + tid] = do_;
      dg_pre[base + tid] = di;
      dg_pre[base + LSTM_H + tid] = df;
      dg_pre[base + 2 * LSTM_H + tid] = dg;
      dg_pre[base + 3 * LSTM_H + tid] = do_;
    }
    __syncthreads();
    float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
    #pragma unroll
    for (int jj = 0; jj < WREG_B; ++jj) {
      float4 dv = dgp4[q * (LSTM_H / 4) + jj];
      s0 = fmaf(wregB[jj].x, dv.x, s0);
      s1 = fmaf(wregB[jj].y, dv.y, s1);
      s2 = fmaf(wregB[jj].z, dv.z, s2);
      s3 = fmaf(wregB[jj].w, dv.w, s3);
    }
    #pragma unroll 16
    for (int jj = WREG_B; jj < LSTM_H / 4; ++jj) {
      int jg = q * (LSTM_H / 4) + jj;    // float4 group of 4 j's
      float4 wv = wpB[(long long)jg * LSTM_H + h];
      float4 dv = dgp4[jg];
      s0 = fmaf(wv.x, dv.x, s0);
      s1 = fmaf(wv.y, dv.y, s1);
      s2 = fmaf(wv.z, dv.z, s2);
      s3 = fmaf(wv.w, dv.w, s3);
    }
    float s = (s0 + s1) + (s2 + s3);
    part[tid] = s;
    __syncthreads();
    if (tid < LSTM_H)
      dh_rec_l[tid] = part[tid] + part[LSTM_H + tid]
                      + part[2 * LSTM_H + tid] + part[3 * LSTM_H + tid];
    __syncthreads();
  }
}

extern "C" void launch_lstm_seq_fwd_b(const float* xp,
                                      const float* w_hh_t_stack,
                                      float* h_seq, float* gates,
                                      float* c_seq, int R,
                                      int rows_per_client, int T,
                                      hipStream_t s) {
  hipLaunchKernelGGL(k_lstm_seq_fwd_b, dim3(R), dim3(4 * LSTM_H), 0, s,
                     xp, w_hh_t_stack, h_seq, gates, c_seq,
                     rows_per_client, T);
}

extern "C" void launch_lstm_seq_bwd_b(const float* gates, const float* c_seq,
                                      const float* w_hh_stack,
                                      const float* dh_out, float* dg_pre,
                                      int R, int rows_per_client, int T,
                                      hipStream_t s) {
  hipLaunchKernelGGL(k_lstm_seq_bwd_b, dim3(R), dim3(4 * LSTM_H), 0, s,
                     gates, c_seq, w_hh_stack, dh_out, dg_pre,
                     rows_per_client, T);
}
