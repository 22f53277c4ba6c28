// Fully-fused CNN-FEMNIST client step for gfx950 (BASELINE north star:
// "Client.process_round() local-SGD ... as hand-written CDNA4 HIP
// kernels").  The flagship benchmark model (reference
// experiments/cv_cnn_femnist/model.py: conv1 1->32 3x3, relu, conv2
// 32->64 3x3, relu, maxpool2, dropout .25, fc1 9216->128, relu,
// dropout .5, fc2 128->C, CE) trained WITHOUT torch autograd: forward,
// backward, clip+stats and the SGD step are ~17 kernel launches per
// batch driven by ONE host call per epoch (launch_cnn_epoch).
//
// Why: hipGraphLaunch costs ~4.6us/node on this stack, so the ~230-node
// autograd capture costs ~1.07 ms host per client epoch (PERF.md).  This
// path is ~85 raw launches (~2us each) per epoch and owns its numerics:
// gradients are written (not accumulated) straight into the flat arena
// at the parameters' fixed offsets, so no zero_grad is needed either.
//
// Shapes are the task's (28x28 in, spatial 26/24/12 fixed); batch B and
// class count C are runtime args.  Dropout uses Philox keyed by
// (seed, batch_index) — deterministic per client epoch.

#include <hip/hip_runtime.h>
#include <hiprand/hiprand_kernel.h>

#define FBLK 256

// arena offsets (fp32 indices) for C classes
struct CnnOffsets {
  long long w1, b1, w2, b2, w3, b3, w4, b4, total;
};

static CnnOffsets cnn_offsets(int C) {
  CnnOffsets o;
  o.w1 = 0;               // [32,1,3,3]
  o.b1 = o.w1 + 288;      // [32]
  o.w2 = o.b1 + 32;       // [64,32,3,3]
  o.b2 = o.w2 + 18432;    // [64]
  o.w3 = o.b2 + 64;       // [128,9216]
  o.b3 = o.w3 + 1179648;  // [128]
  o.w4 = o.b3 + 128;      // [C,128]
  o.b4 = o.w4 + (long long)C * 128;  // [C]
  o.total = o.b4 + C;
  return o;
}

// ---------------------------------------------------------------------------
// gather a shuffled batch from the device-resident shard
// ---------------------------------------------------------------------------
__global__ void k_gather_batch(const float* __restrict__ shard_x,
                               const long long* __restrict__ shard_y,
                               const long long* __restrict__ order,
                               long long start, long long row_base, int B,
                               float* __restrict__ xb,
                               int* __restrict__ yb) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < B * 784;
       i += gridDim.x * blockDim.x) {
    int b = i / 784, j = i % 784;
    long long src = row_base + order[start + b];
    xb[i] = shard_x[src * 784 + j];
    if (j == 0) yb[b] = (int)shard_y[src];
  }
}

// grad = (w_server - w_trained) * weight  (client pseudo-gradient, K1+K2)
__global__ void k_cnn_pseudo_grad(float* __restrict__ g,
                                  const float* __restrict__ ws,
                                  const float* __restrict__ wt, float weight,
                                  long long n) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x)
    g[i] = (ws[i] - wt[i]) * weight;
}

__global__ void k_cnn_axpy(float* __restrict__ y, const float* __restrict__ x,
                           long long n) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x)
    y[i] += x[i];
}

__global__ void k_copy(float* __restrict__ dst, const float* __restrict__ src,
                       long long n) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x)
    dst[i] = src[i];
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
// a1[b,co,y,x] = relu(b1 + sum_{kh,kw} w1[co,kh,kw] * x[b, y+kh, x+kw])
__global__ void k_conv1_fwd(const float* __restrict__ x,
                            const float* __restrict__ w1,
                            const float* __restrict__ b1, int B,
                            float* __restrict__ a1) {
  int total = B * 32 * 26 * 26;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int xx = i % 26, yy = (i / 26) % 26, co = (i / 676) % 32, b = i / 21632;
    const float* xp = x + b * 784 + yy * 28 + xx;
    const float* wp = w1 + co * 9;
    float acc = b1[co];
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh)
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw)
        acc = fmaf(wp[kh * 3 + kw], xp[kh * 28 + kw], acc);
    a1[i] = acc > 0.f ? acc : 0.f;
  }
}

// r2[b,co,y,x] = relu(b2 + sum_ci sum_k w2[co,ci,k]*a1[b,ci,y+kh,x+kw])
// One block per (b, group of 8 co); a1's 32 input channels are staged
// through LDS in two 16-channel tiles (42.25 KB) and reused by all 8
// output channels — 8x less LDS-staging traffic than one-co-per-block.
// W2 taps for the tile live in registers (8 co x 9 = 72 floats/thread is
// too many, so taps reload per ci from L1 — W2 is 72 KB, L2-resident).
#define CONV2_COG 8
__global__ void k_conv2_fwd(const float* __restrict__ a1,
                            const float* __restrict__ w2,
                            const float* __restrict__ b2, int B,
                            float* __restrict__ r2) {
  __shared__ float lds[16 * 676];
  int b = blockIdx.x / (64 / CONV2_COG);
  int co0 = (blockIdx.x % (64 / CONV2_COG)) * CONV2_COG;
  float acc[3][CONV2_COG];
  #pragma unroll
  for (int r = 0; r < 3; ++r)
    #pragma unroll
    for (int g = 0; g < CONV2_COG; ++g) acc[r][g] = b2[co0 + g];
  for (int half = 0; half < 2; ++half) {
    const float* src = a1 + ((long long)b * 32 + half * 16) * 676;
    __syncthreads();
    for (int i = threadIdx.x; i < 16 * 676; i += blockDim.x)
      lds[i] = src[i];
    __syncthreads();
    #pragma unroll
    for (int r = 0; r < 3; ++r) {
      int o = threadIdx.x + r * FBLK;
      if (o >= 576) continue;
      int xx = o % 24, yy = o / 24;
      for (int ci = 0; ci < 16; ++ci) {
        const float* ap = lds + ci * 676 + yy * 26 + xx;
        float a[9];
        #pragma unroll
        for (int kh = 0; kh < 3; ++kh)
          #pragma unroll
          for (int kw = 0; kw < 3; ++kw)
            a[kh * 3 + kw] = ap[kh * 26 + kw];
        #pragma unroll
        for (int g = 0; g < CONV2_COG; ++g) {
          const float* wq = w2 + ((long long)(co0 + g) * 32
                                  + half * 16 + ci) * 9;
          float s = acc[r][g];
          #pragma unroll
          for (int k = 0; k < 9; ++k) s = fmaf(wq[k], a[k], s);
          acc[r][g] = s;
        }
      }
    }
  }
  #pragma unroll
  for (int r = 0; r < 3; ++r) {
    int o = threadIdx.x + r * FBLK;
    if (o < 576)
      #pragma unroll
      for (int g = 0; g < CONV2_COG; ++g) {
        float v = acc[r][g];
        r2[((long long)b * 64 + co0 + g) * 576 + o] = v > 0.f ? v : 0.f;
      }
  }
}

// maxpool 2x2 + dropout(p1): a2 = keep ? max/(1-p1) : 0; save argmax+mask
__global__ void k_pool_drop_fwd(const float* __restrict__ r2, int B,
                                float p1, unsigned long long seed,
                                unsigned long long offset,
                                float* __restrict__ a2,
                                unsigned char* __restrict__ pidx,
                                unsigned char* __restrict__ m2) {
  int total = B * 64 * 144;
  float inv_keep = (p1 < 1.f) ? 1.f / (1.f - p1) : 0.f;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int px = i % 12, py = (i / 12) % 12, c = (i / 144) % 64, b = i / 9216;
    const float* base = r2 + (((long long)b * 64 + c) * 24 + 2 * py) * 24 + 2 * px;
    float v0 = base[0], v1 = base[1], v2 = base[24], v3 = base[25];
    float m = v0; int idx = 0;
    if (v1 > m) { m = v1; idx = 1; }
    if (v2 > m) { m = v2; idx = 2; }
    if (v3 > m) { m = v3; idx = 3; }
    unsigned char keep = 1;
    if (p1 > 0.f) {
      // one Philox draw covers 4 cells (init is ~10 rounds — the cost)
      hiprandStatePhilox4_32_10_t st;
      hiprand_init(seed, (unsigned long long)(i >> 2), offset, &st);
      float4 u = hiprand_uniform4(&st);
      float uu = (i & 3) == 0 ? u.x : (i & 3) == 1 ? u.y
                 : (i & 3) == 2 ? u.z : u.w;
      keep = uu >= p1;
    }
    pidx[i] = (unsigned char)idx;
    m2[i] = keep;
    a2[i] = keep ? m * inv_keep : 0.f;
  }
}

// z3 = W3 @ a2flat + b3 ; a3 = dropout(relu(z3), p2)
// one block per (b, j): 256-thread coalesced dot of length 9216.  (A
// one-block-per-j variant amortizing W3 reads across B measured 3x
// slower: the per-b a2 reads stride 9216 and kill coalescing; here both
// streams are contiguous and L2 absorbs the W3 re-reads.)
__global__ void k_fc1_fwd(const float* __restrict__ a2,
                          const float* __restrict__ w3,
                          const float* __restrict__ b3, int B, float p2,
                          unsigned long long seed, unsigned long long offset,
                          float* __restrict__ z3, float* __restrict__ a3,
                          unsigned char* __restrict__ m3) {
  int b = blockIdx.x / 128, j = blockIdx.x % 128;
  const float* ap = a2 + (long long)b * 9216;
  const float* wp = w3 + (long long)j * 9216;
  float s = 0.f;
  for (int k = threadIdx.x; k < 9216; k += blockDim.x)
    s = fmaf(wp[k], ap[k], s);
  for (int d = 32; d > 0; d >>= 1) s += __shfl_down(s, d, 64);
  __shared__ float lds[FBLK / 64];
  if ((threadIdx.x & 63) == 0) lds[threadIdx.x >> 6] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = b3[j];
    for (int w = 0; w < FBLK / 64; ++w) t += lds[w];
    int i = b * 128 + j;
    z3[i] = t;
    float r = t > 0.f ? t : 0.f;
    unsigned char keep = 1;
    if (p2 > 0.f) {
      hiprandStatePhilox4_32_10_t st;
      hiprand_init(seed ^ 0x9e3779b97f4a7c15ull, (unsigned long long)i,
                   offset, &st);
      keep = hiprand_uniform(&st) >= p2;
    }
    m3[i] = keep;
    a3[i] = keep ? r / (1.f - p2) : 0.f;
  }
}

// logits = W4 @ a3 + b4 ; softmax ; loss += -log p[target]/B ;
// dlogit = (p - onehot)/B.  One block per sample row.
__global__ void k_fc2_loss_fwd(const float* __restrict__ a3,
                               const float* __restrict__ w4,
                               const float* __restrict__ b4,
                               const int* __restrict__ yb, int B, int C,
                               float* __restrict__ dlogits,
                               float* __restrict__ loss_acc) {
  extern __shared__ float sm[];  // C floats
  int b = blockIdx.x;
  const float* ap = a3 + (long long)b * 128;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    const float* wp = w4 + (long long)j * 128;
    float s = b4[j];
    #pragma unroll 4
    for (int k = 0; k < 128; ++k) s = fmaf(wp[k], ap[k], s);
    sm[j] = s;
  }
  __syncthreads();
  // softmax over C (single wave handles it: C <= 1024 assumed small)
  if (threadIdx.x == 0) {
    float mx = sm[0];
    for (int j = 1; j < C; ++j) mx = fmaxf(mx, sm[j]);
    float z = 0.f;
    for (int j = 0; j < C; ++j) { sm[j] = __expf(sm[j] - mx); z += sm[j]; }
    float inv = 1.f / z;
    int t = yb[b];
    for (int j = 0; j < C; ++j) {
      float p = sm[j] * inv;
      dlogits[(long long)b * C + j] = (p - (j == t ? 1.f : 0.f)) / (float)B;
    }
    atomicAdd(loss_acc, -__logf(fmaxf(sm[t] * inv, 1e-30f)) / (float)B);
  }
}

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------
// dW4[j,k] = sum_b dlogits[b,j]*a3[b,k]; db4[j] = sum_b dlogits[b,j]
__global__ void k_fc2_bwd_w(const float* __restrict__ dlogits,
                            const float* __restrict__ a3, int B, int C,
                            float* __restrict__ dw4, float* __restrict__ db4) {
  int total = C * 128;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int k = i % 128, j = i / 128;
    float s = 0.f, sb = 0.f;
    for (int b = 0; b < B; ++b) {
      float d = dlogits[(long long)b * C + j];
      s = fmaf(d, a3[(long long)b * 128 + k], s);
      if (k == 0) sb += d;
    }
    dw4[i] = s;
    if (k == 0) db4[j] = sb;
  }
}

// dz3[b,k] = (sum_j dlogits[b,j]*W4[j,k]) * m3/(1-p2) * (z3>0)
__global__ void k_fc2_bwd_x(const float* __restrict__ dlogits,
                            const float* __restrict__ w4,
                            const float* __restrict__ z3,
                            const unsigned char* __restrict__ m3, int B,
                            int C, float p2, float* __restrict__ dz3) {
  int total = B * 128;
  float inv_keep = 1.f / (1.f - p2);
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int k = i % 128, b = i / 128;
    float s = 0.f;
    for (int j = 0; j < C; ++j)
      s = fmaf(dlogits[(long long)b * C + j], w4[(long long)j * 128 + k], s);
    float g = (p2 > 0.f) ? (m3[i] ? s * inv_keep : 0.f) : s;
    dz3[i] = z3[i] > 0.f ? g : 0.f;
  }
}

// dW3[j,k] = sum_b dz3[b,j]*a2[b,k]; db3[j] = sum_b dz3[b,j]
__global__ void k_fc1_bwd_w(const float* __restrict__ dz3,
                            const float* __restrict__ a2, int B,
                            float* __restrict__ dw3, float* __restrict__ db3) {
  long long total = 128LL * 9216;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i % 9216), j = (int)(i / 9216);
    float s = 0.f, sb = 0.f;
    for (int b = 0; b < B; ++b) {
      float d = dz3[b * 128 + j];
      s = fmaf(d, a2[(long long)b * 9216 + k], s);
      if (k == 0) sb += d;
    }
    dw3[i] = s;
    if (k == 0) db3[j] = sb;
  }
}

// d_a2[b,k] = sum_j dz3[b,j] * W3[j,k]  (then dropout1 bwd is fused in
// the pool scatter)
__global__ void k_fc1_bwd_x(const float* __restrict__ dz3,
                            const float* __restrict__ w3, int B,
                            float* __restrict__ da2) {
  long long total = (long long)B * 9216;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i % 9216), b = (int)(i / 9216);
    const float* dp = dz3 + b * 128;
    float s = 0.f;
    #pragma unroll 4
    for (int j = 0; j < 128; ++j)
      s = fmaf(dp[j], w3[(long long)j * 9216 + k], s);
    da2[i] = s;
  }
}

// dropout1 bwd + maxpool scatter + relu bwd: one thread per pool cell
// writes its 2x2 input cells (exactly one gets the grad)
__global__ void k_pool_drop_bwd(const float* __restrict__ da2,
                                const unsigned char* __restrict__ pidx,
                                const unsigned char* __restrict__ m2,
                                const float* __restrict__ r2, int B, float p1,
                                float* __restrict__ dz2) {
  int total = B * 64 * 144;
  float inv_keep = (p1 < 1.f) ? 1.f / (1.f - p1) : 0.f;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int px = i % 12, py = (i / 12) % 12, c = (i / 144) % 64, b = i / 9216;
    float g = (p1 > 0.f) ? (m2[i] ? da2[i] * inv_keep : 0.f) : da2[i];
    int idx = pidx[i];
    long long base = (((long long)b * 64 + c) * 24 + 2 * py) * 24 + 2 * px;
    #pragma unroll
    for (int d = 0; d < 4; ++d) {
      long long o = base + (d >> 1) * 24 + (d & 1);
      float v = (d == idx && r2[o] > 0.f) ? g : 0.f;
      dz2[o] = v;
    }
  }
}

// dW2[co,ci,kh,kw] = sum_{b,y,x} a1[b,ci,y+kh,x+kw] * dz2[b,co,y,x]
// one block per (co,ci) pair: 9 kernel taps reduced across 256 threads
__global__ void k_conv2_bwd_w(const float* __restrict__ a1,
                              const float* __restrict__ dz2, int B,
                              float* __restrict__ dw2,
                              float* __restrict__ db2) {
  int co = blockIdx.x / 32, ci = blockIdx.x % 32;
  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float accb = 0.f;
  for (int t = threadIdx.x; t < B * 576; t += blockDim.x) {
    int o = t % 576, b = t / 576;
    int xx = o % 24, yy = o / 24;
    float d = dz2[((long long)b * 64 + co) * 576 + o];
    const float* ap = a1 + ((long long)b * 32 + ci) * 676 + yy * 26 + xx;
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh)
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw)
        acc[kh * 3 + kw] = fmaf(ap[kh * 26 + kw], d, acc[kh * 3 + kw]);
    if (ci == 0) accb += d;
  }
  __shared__ float lds[FBLK / 64 * 10];
  int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  #pragma unroll
  for (int k = 0; k < 9; ++k) {
    float s = acc[k];
    for (int d = 32; d > 0; d >>= 1) s += __shfl_down(s, d, 64);
    if (lane == 0) lds[wave * 10 + k] = s;
  }
  float sb = accb;
  for (int d = 32; d > 0; d >>= 1) sb += __shfl_down(sb, d, 64);
  if (lane == 0) lds[wave * 10 + 9] = sb;
  __syncthreads();
  if (threadIdx.x < 9) {
    float s = 0.f;
    for (int w = 0; w < FBLK / 64; ++w) s += lds[w * 10 + threadIdx.x];
    dw2[(long long)(co * 32 + ci) * 9 + threadIdx.x] = s;
  }
  if (threadIdx.x == 9 && ci == 0) {
    float s = 0.f;
    for (int w = 0; w < FBLK / 64; ++w) s += lds[w * 10 + 9];
    db2[co] = s;
  }
}

// dz1[b,ci,p,q] = relu'(a1) * sum_{co,kh,kw valid} W2[co,ci,kh,kw] *
//                 dz2[b,co,p-kh,q-kw]
// Flat one-thread-per-output form: the LDS-tiled variant measured SLOWER
// (160 blocks underfill the 256-CU chip and serialize on barriers); here
// W2 (72 KB) and the dz2 rows stream through L2 instead.
__global__ void k_conv2_bwd_x(const float* __restrict__ dz2,
                              const float* __restrict__ w2,
                              const float* __restrict__ a1, int B,
                              float* __restrict__ dz1) {
  int total = B * 32 * 676;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int q = i % 26, p = (i / 26) % 26, ci = (i / 676) % 32, b = i / 21632;
    if (a1[i] <= 0.f) { dz1[i] = 0.f; continue; }
    float s = 0.f;
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      int y = p - kh;
      if (y < 0 || y >= 24) continue;
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        int x = q - kw;
        if (x < 0 || x >= 24) continue;
        for (int co = 0; co < 64; ++co)
          s = fmaf(w2[((long long)co * 32 + ci) * 9 + kh * 3 + kw],
                   dz2[((long long)b * 64 + co) * 576 + y * 24 + x], s);
      }
    }
    dz1[i] = s;
  }
}

// dW1[co,kh,kw] = sum_{b,y,x} x[b,y+kh,x+kw] * dz1[b,co,y,x]; db1 likewise
__global__ void k_conv1_bwd_w(const float* __restrict__ x,
                              const float* __restrict__ dz1, int B,
                              float* __restrict__ dw1,
                              float* __restrict__ db1) {
  int co = blockIdx.x;  // 32 blocks
  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float accb = 0.f;
  for (int t = threadIdx.x; t < B * 676; t += blockDim.x) {
    int o = t % 676, b = t / 676;
    int xx = o % 26, yy = o / 26;
    float d = dz1[((long long)b * 32 + co) * 676 + o];
    const float* xp = x + b * 784 + yy * 28 + xx;
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh)
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw)
        acc[kh * 3 + kw] = fmaf(xp[kh * 28 + kw], d, acc[kh * 3 + kw]);
    accb += d;
  }
  __shared__ float lds[16 * 10];  // up to 16 waves (1024-thread blocks)
  int n_waves = blockDim.x / 64;
  int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  #pragma unroll
  for (int k = 0; k < 9; ++k) {
    float s = acc[k];
    for (int d = 32; d > 0; d >>= 1) s += __shfl_down(s, d, 64);
    if (lane == 0) lds[wave * 10 + k] = s;
  }
  float sb = accb;
  for (int d = 32; d > 0; d >>= 1) sb += __shfl_down(sb, d, 64);
  if (lane == 0) lds[wave * 10 + 9] = sb;
  __syncthreads();
  if (threadIdx.x < 9) {
    float s = 0.f;
    for (int w = 0; w < n_waves; ++w) s += lds[w * 10 + threadIdx.x];
    dw1[co * 9 + threadIdx.x] = s;
  }
  if (threadIdx.x == 9) {
    float s = 0.f;
    for (int w = 0; w < n_waves; ++w) s += lds[w * 10 + 9];
    db1[co] = s;
  }
}

// ---------------------------------------------------------------------------
// MFMA implicit-GEMM forms of the profile-dominant kernels (round-1
// profile: k_fc1_fwd 18.3%, k_conv2_bwd_x 18.3%, k_conv2_fwd 16.6%,
// k_conv2_bwd_w 12.5% — profiles/r01_bench_kernel_stats_final.md).
// gfx950's f32-input MFMA (v_mfma_f32_16x16x4_f32) is EXACT f32 — a
// k-ordered fmaf chain at the full f32 rate — so these keep fp32
// numerics while moving the matrix work onto the matrix pipe and leaving
// the VALU free for address math and epilogues.
//
// Fragment maps (guide §3): A[i=l&15][k=l>>4], B[k=l>>4][j=l&15] one f32
// VGPR each; D[row=(l>>4)*4+r][col=l&15], 4 f32 per lane.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(4))) float f32x4;

// W2 fragment layouts, rebuilt once per batch (the SGD step changes W2):
// w2t[k=(ci,kh,kw)][co]   — conv2-fwd B operand, lane-contiguous in co
// w2rot[k=(co,kh,kw)][ci] — conv2-bwd-data B operand, lane-contiguous in ci
__global__ void k_w2_layouts(const float* __restrict__ w2,
                             float* __restrict__ w2t,
                             float* __restrict__ w2rot) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= 18432) return;
  int co = i / 288, k = i % 288;
  float v = w2[i];
  w2t[k * 64 + co] = v;
  int ci = k / 9, rem = k % 9;
  w2rot[(co * 9 + rem) * 32 + ci] = v;
}

// dz2 transposed to [b][o][co] (written into ws.r2, which is free after
// pool_drop_bwd) — conv2-bwd-weight A operand, lane-contiguous in co
__global__ void k_dz2_transpose(const float* __restrict__ dz2, int B,
                                float* __restrict__ dz2t) {
  int total = B * 36864;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    int b = i / 36864, r = i % 36864, o = r / 64, co = r % 64;
    dz2t[i] = dz2[((long long)b * 64 + co) * 576 + o];
  }
}

// conv2 forward as implicit GEMM: out[(b,o),co] = sum_k a1im2col[(b,o),k]
// * W2^T[k,co], M=576/batch-row (exact 36 16-tiles), N=64, K=288.
// One block per (b, 64-row M-tile); a1[b] (86.5 KB) staged once in LDS.
__global__ __launch_bounds__(256)
void k_conv2_fwd_mfma(const float* __restrict__ a1,
                      const float* __restrict__ w2t,
                      const float* __restrict__ b2, int B,
                      float* __restrict__ r2) {
  __shared__ float lds[32 * 676];
  int b = blockIdx.x / 9, mt = blockIdx.x % 9;
  const float* src = a1 + (long long)b * 21632;
  for (int i = threadIdx.x; i < 21632; i += 256) lds[i] = src[i];
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int m = mt * 64 + w * 16 + (lane & 15);  // output cell o in [0,576)
  int yy = m / 24, xx = m % 24;
  int kc = lane >> 4;
  f32x4 acc[4] = {f32x4{0,0,0,0}, f32x4{0,0,0,0},
                  f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 288; k0 += 4) {
    int k = k0 + kc;
    int ci = k / 9, rem = k % 9, kh = rem / 3, kw = rem % 3;
    float a = lds[ci * 676 + (yy + kh) * 26 + xx + kw];
    const float* wrow = w2t + (long long)k * 64 + (lane & 15);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      float bv = wrow[nt * 16];  // lane-contiguous 64B group reads
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[nt], 0, 0, 0);
    }
  }
  int om = mt * 64 + w * 16 + (lane >> 4) * 4;
  int cl = lane & 15;
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      int co = nt * 16 + cl;
      float v = acc[nt][r] + b2[co];
      r2[((long long)b * 64 + co) * 576 + om + r] = v > 0.f ? v : 0.f;
    }
}

// conv2 backward-data as implicit GEMM: dz1[(b,p,q),ci] = relu'(a1) *
// sum_k dz2pad[(b,p,q),k] * W2rot[k,ci], M=676/batch-row (11 64-tiles,
// masked tail), N=32, K=576 (co,kh,kw).  dz2[b] (147 KB) staged in LDS.
__global__ __launch_bounds__(256)
void k_conv2_bwd_x_mfma(const float* __restrict__ dz2,
                        const float* __restrict__ w2rot,
                        const float* __restrict__ a1, int B,
                        float* __restrict__ dz1) {
  __shared__ float lds[64 * 576];
  int b = blockIdx.x / 11, mt = blockIdx.x % 11;
  const float* src = dz2 + (long long)b * 36864;
  for (int i = threadIdx.x; i < 36864; i += 256) lds[i] = src[i];
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int m = mt * 64 + w * 16 + (lane & 15);  // cell (p,q) in [0,676)
  int p = m / 26, q = m % 26;
  bool mrow = m < 676;
  int kc = lane >> 4;
  f32x4 acc[2] = {f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 576; k0 += 4) {
    int k = k0 + kc;
    int co = k / 9, rem = k % 9, kh = rem / 3, kw = rem % 3;
    int y = p - kh, x = q - kw;
    float a = (mrow && y >= 0 && y < 24 && x >= 0 && x < 24)
                  ? lds[co * 576 + y * 24 + x] : 0.f;
    const float* wrow = w2rot + (long long)k * 32 + (lane & 15);
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      float bv = wrow[nt * 16];  // lane-contiguous 64B group reads
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[nt], 0, 0, 0);
    }
  }
  int om = mt * 64 + w * 16 + (lane >> 4) * 4;
  int cl = lane & 15;
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    if (om + r >= 676) continue;
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      int ci = nt * 16 + cl;
      long long o = ((long long)b * 32 + ci) * 676 + om + r;
      dz1[o] = a1[o] > 0.f ? acc[nt][r] : 0.f;
    }
  }
}

// conv2 weight gradient as GEMM: dW2[co,(ci,kh,kw)] = sum_{b,o}
// dz2[(b,o),co] * a1im2col[(b,o),(ci,kh,kw)], M=64, N=288, K=B*576,
// split over b: block (nb in 0..5, b) computes a 64x48 tile of the
// per-b partial into ws.wsl[b]; k_conv2_bwd_w_fold sums the B slabs in
// fixed order (deterministic — no atomics).  a1[b] staged in LDS.
__global__ __launch_bounds__(256)
void k_conv2_bwd_w_mfma(const float* __restrict__ dz2t,
                        const float* __restrict__ a1, int B,
                        float* __restrict__ slab) {
  __shared__ float lds[32 * 676];
  int nb = blockIdx.x % 6, b = blockIdx.x / 6;
  const float* src = a1 + (long long)b * 21632;
  for (int i = threadIdx.x; i < 21632; i += 256) lds[i] = src[i];
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int co = w * 16 + (lane & 15);
  int kc = lane >> 4;
  const float* dzb = dz2t + (long long)b * 36864;  // [o][co] layout
  f32x4 acc[3] = {f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 576; k0 += 4) {
    int o = k0 + kc;
    int yy = o / 24, xx = o % 24;
    float a = dzb[o * 64 + co];  // lane-contiguous 64B group reads
    #pragma unroll
    for (int nt = 0; nt < 3; ++nt) {
      int n = nb * 48 + nt * 16 + (lane & 15);
      int ci = n / 9, rem = n % 9, kh = rem / 3, kw = rem % 3;
      float bv = lds[ci * 676 + (yy + kh) * 26 + xx + kw];
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[nt], 0, 0, 0);
    }
  }
  float* out = slab + (long long)b * 18432;
  int orow = w * 16 + (lane >> 4) * 4;
  int cl = lane & 15;
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    #pragma unroll
    for (int nt = 0; nt < 3; ++nt)
      out[(orow + r) * 288 + nb * 48 + nt * 16 + cl] = acc[nt][r];
}

// fold the B per-batch-row dW2 slabs (fixed order) + db2 column sums
__global__ void k_conv2_bwd_w_fold(const float* __restrict__ slab, int B,
                                   float* __restrict__ dw2) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= 18432) return;
  float s = 0.f;
  for (int b = 0; b < B; ++b) s += slab[(long long)b * 18432 + i];
  dw2[i] = s;
}

__global__ void k_conv2_bwd_b(const float* __restrict__ dz2, int B,
                              float* __restrict__ db2) {
  int co = blockIdx.x;
  float s = 0.f;
  for (int t = threadIdx.x; t < B * 576; t += blockDim.x) {
    int o = t % 576, b = t / 576;
    s += dz2[((long long)b * 64 + co) * 576 + o];
  }
  for (int d = 32; d > 0; d >>= 1) s += __shfl_down(s, d, 64);
  __shared__ float lds[4];
  if ((threadIdx.x & 63) == 0) lds[threadIdx.x >> 6] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int k = 0; k < (int)blockDim.x / 64; ++k) t += lds[k];
    db2[co] = t;
  }
}

// fc1 forward as split-K MFMA GEMM: z3^T[j,b] = sum_k W3[j,k]*a2[b,k],
// M=128 (j), N=32 (b, masked to B), K=9216 split into FC1_SPLIT chunks;
// per-chunk partials land in ws.wsl and k_fc1_fwd_reduce folds them in
// fixed order + bias + relu + dropout (same Philox stream as the scalar
// k_fc1_fwd, so masks are bit-identical to round 1's kernel).
#define FC1_SPLIT 64
#define FC1_CH (9216 / FC1_SPLIT)  /* 144 k per chunk */
#define FC1_LD (FC1_CH + 1)        /* LDS row stride 145: odd multiplier of
                                      16 mod 32 -> conflict-free group reads */
__global__ __launch_bounds__(256)
void k_fc1_fwd_mfma(const float* __restrict__ a2,
                    const float* __restrict__ w3, int B,
                    float* __restrict__ slab) {
  __shared__ float lw[128 * FC1_LD];  // W3 chunk [128][144] (72.5 KB)
  __shared__ float la[32 * FC1_LD];   // a2 chunk [32][144]  (18.1 KB)
  int s = blockIdx.x;                 // k-chunk
  int k_base = s * FC1_CH;
  for (int i = threadIdx.x; i < 128 * FC1_CH; i += 256) {
    int row = i / FC1_CH, kk = i % FC1_CH;  // 576 B coalesced runs per row
    lw[row * FC1_LD + kk] = w3[(long long)row * 9216 + k_base + kk];
  }
  for (int i = threadIdx.x; i < 32 * FC1_CH; i += 256) {
    int bu = i / FC1_CH, kk = i % FC1_CH;
    la[bu * FC1_LD + kk] = bu < B ? a2[(long long)bu * 9216 + k_base + kk]
                                  : 0.f;
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int kc = lane >> 4, il = lane & 15;
  f32x4 acc[2][2] = {{f32x4{0,0,0,0}, f32x4{0,0,0,0}},
                     {f32x4{0,0,0,0}, f32x4{0,0,0,0}}};
  for (int k0 = 0; k0 < FC1_CH; k0 += 4) {
    int k = k0 + kc;
    float a0 = lw[(w * 32 + il) * FC1_LD + k];
    float a1v = lw[(w * 32 + 16 + il) * FC1_LD + k];
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      float bv = la[(u * 16 + il) * FC1_LD + k];
      acc[0][u] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, bv, acc[0][u], 0, 0, 0);
      acc[1][u] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1v, bv, acc[1][u], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int t = 0; t < 2; ++t)
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      int bu = u * 16 + il;
      if (bu >= B) continue;
      int j = w * 32 + t * 16 + (lane >> 4) * 4;
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        slab[((long long)s * B + bu) * 128 + j + r] = acc[t][u][r];
    }
}

__global__ void k_fc1_fwd_reduce(const float* __restrict__ slab,
                                 const float* __restrict__ b3, int B,
                                 float p2, unsigned long long seed,
                                 unsigned long long offset,
                                 float* __restrict__ z3,
                                 float* __restrict__ a3,
                                 unsigned char* __restrict__ m3) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B * 128) return;
  int b = i / 128, j = i % 128;
  float t = b3[j];
  for (int s = 0; s < FC1_SPLIT; ++s)
    t += slab[((long long)s * B + b) * 128 + j];
  z3[i] = t;
  float r = t > 0.f ? t : 0.f;
  unsigned char keep = 1;
  if (p2 > 0.f) {
    hiprandStatePhilox4_32_10_t st;
    hiprand_init(seed ^ 0x9e3779b97f4a7c15ull, (unsigned long long)i,
                 offset, &st);
    keep = hiprand_uniform(&st) >= p2;
  }
  m3[i] = keep;
  a3[i] = keep ? r / (1.f - p2) : 0.f;
}

// dW3[j,n] = sum_b dz3[b,j]*a2[b,n]: M=128, N=9216, K=B (<=32, masked).
// Block = 128x64 tile (wave: 2 m-tiles x 4 n-tiles), grid = 144.
__global__ __launch_bounds__(256)
void k_fc1_bwd_w_mfma(const float* __restrict__ dz3,
                      const float* __restrict__ a2, int B,
                      float* __restrict__ dw3) {
  int nblk = blockIdx.x;                  // 64-col slab of N
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int kc = lane >> 4, il = lane & 15;
  f32x4 acc[2][4] = {{f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}},
                     {f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}}};
  for (int k0 = 0; k0 < ((B + 3) & ~3); k0 += 4) {
    int b = k0 + kc;
    bool kv = b < B;
    float a0 = kv ? dz3[(long long)b * 128 + w * 32 + il] : 0.f;
    float a1v = kv ? dz3[(long long)b * 128 + w * 32 + 16 + il] : 0.f;
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      long long n = (long long)nblk * 64 + nt * 16 + il;
      float bv = kv ? a2[(long long)b * 9216 + n] : 0.f;
      acc[0][nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, bv, acc[0][nt], 0, 0, 0);
      acc[1][nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1v, bv, acc[1][nt], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int t = 0; t < 2; ++t) {
    int j = w * 32 + t * 16 + (lane >> 4) * 4;
    #pragma unroll
    for (int r = 0; r < 4; ++r)
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        dw3[(long long)(j + r) * 9216 + nblk * 64 + nt * 16 + il] =
            acc[t][nt][r];
  }
}

__global__ void k_fc1_bwd_b(const float* __restrict__ dz3, int B,
                            float* __restrict__ db3) {
  int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= 128) return;
  float s = 0.f;
  for (int b = 0; b < B; ++b) s += dz3[(long long)b * 128 + j];
  db3[j] = s;
}

// da2[b,n] = sum_j dz3[b,j]*W3[j,n]: M=32 (b, masked to B), N=9216,
// K=128.  Block = 32x64 tile (wave: 1 n-tile of 16 cols x 2 m-tiles),
// grid = 144.
__global__ __launch_bounds__(256)
void k_fc1_bwd_x_mfma(const float* __restrict__ dz3,
                      const float* __restrict__ w3, int B,
                      float* __restrict__ da2) {
  int nblk = blockIdx.x;
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int kc = lane >> 4, il = lane & 15;
  f32x4 acc[2] = {f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 128; k0 += 4) {
    int j = k0 + kc;
    long long n = (long long)nblk * 64 + w * 16 + il;
    float bv = w3[(long long)j * 9216 + n];
    #pragma unroll
    for (int t = 0; t < 2; ++t) {
      int bu = t * 16 + il;
      float a = bu < B ? dz3[(long long)bu * 128 + j] : 0.f;
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[t], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int t = 0; t < 2; ++t) {
    int bu = t * 16 + (lane >> 4) * 4;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (bu + r >= B) continue;
      da2[(long long)(bu + r) * 9216 + (long long)nblk * 64 + w * 16 + il] =
          acc[t][r];
    }
  }
}

// ---------------------------------------------------------------------------
// epoch driver: the ONLY host entry — loops every batch of one client's
// local epoch, launching fwd/bwd + fused clip-stats + SGD per batch.
// ---------------------------------------------------------------------------

extern "C" {
void launch_sum_sumsq2(const float*, long long, double*, double*, hipStream_t);
void launch_clip_apply_stats(float*, long long, const double*, float, float,
                             float*, hipStream_t);
void launch_sgd_step(float*, const float*, float*, float, const float*, float,
                     float, float, int, int, long long, hipStream_t);
}

struct CnnWorkspace {
  // laid out inside one float buffer by the binding (sizes for B rows)
  float *xb, *a1, *r2, *a2, *z3, *a3, *dlogits, *dz3, *da2, *dz2, *dz1;
  float *wsl;  // split-K partial slab (B*18432 floats), shared fwd/bwd
  float *w2t, *w2rot;  // per-batch W2 fragment layouts (18432 floats each)
  int *yb;
  unsigned char *pidx, *m2, *m3;
  double *red_partials, *red_acc;
};

// ---------------------------------------------------------------------------
// bf16 GEMM kernels (mixed precision: bf16 MFMA inputs at ~13-16x the
// f32 MFMA rate, fp32 accumulators, fp32 master weights/outputs — the
// SGD step, clip/stats, DP/quant hooks all stay fp32).  The GEMM
// operands are materialized in LDS as bf16 with k-contiguous rows so
// each lane's 8-element fragment is ONE 16 B ds_read; row strides are
// padded so 16-lane fragment groups hit distinct banks.
// Conversion f32->bf16 happens during LDS staging (round-to-nearest via
// __bf16 cast) — no global bf16 shadow copies needed.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

__device__ __forceinline__ bf16x8 ld_bf16x8(const __bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

// fc1 forward, bf16: z3^T[j,b] partials; split-K 36 (chunk 256 = 8 bf16
// k-steps).  LDS: W3 chunk [128][256] + a2 chunk [32][256], stride 264.
#define FC1B_SPLIT 36
#define FC1B_CH 256
#define FC1B_LD 264
__global__ __launch_bounds__(256)
void k_fc1_fwd_mfma_bf16(const float* __restrict__ a2,
                         const float* __restrict__ w3, int B,
                         float* __restrict__ slab) {
  __shared__ __bf16 lw[128 * FC1B_LD];
  __shared__ __bf16 la[32 * FC1B_LD];
  int s = blockIdx.x;
  int k_base = s * FC1B_CH;
  for (int i = threadIdx.x; i < 128 * FC1B_CH; i += 256) {
    int row = i / FC1B_CH, kk = i % FC1B_CH;
    lw[row * FC1B_LD + kk] = (__bf16)w3[(long long)row * 9216 + k_base + kk];
  }
  for (int i = threadIdx.x; i < 32 * FC1B_CH; i += 256) {
    int bu = i / FC1B_CH, kk = i % FC1B_CH;
    la[bu * FC1B_LD + kk] =
        (__bf16)(bu < B ? a2[(long long)bu * 9216 + k_base + kk] : 0.f);
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  f32x4 acc[2][2] = {{f32x4{0,0,0,0}, f32x4{0,0,0,0}},
                     {f32x4{0,0,0,0}, f32x4{0,0,0,0}}};
  for (int k0 = 0; k0 < FC1B_CH; k0 += 32) {
    bf16x8 a0 = ld_bf16x8(&lw[(w * 32 + il) * FC1B_LD + k0 + kc8]);
    bf16x8 a1v = ld_bf16x8(&lw[(w * 32 + 16 + il) * FC1B_LD + k0 + kc8]);
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      bf16x8 bv = ld_bf16x8(&la[(u * 16 + il) * FC1B_LD + k0 + kc8]);
      acc[0][u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bv, acc[0][u], 0, 0, 0);
      acc[1][u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1v, bv, acc[1][u], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int t = 0; t < 2; ++t)
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      int bu = u * 16 + il;
      if (bu >= B) continue;
      int j = w * 32 + t * 16 + (lane >> 4) * 4;
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        slab[((long long)s * B + bu) * 128 + j + r] = acc[t][u][r];
    }
}

__global__ void k_fc1_fwd_reduce_bf16(const float* __restrict__ slab,
                                      const float* __restrict__ b3, int B,
                                      float p2, unsigned long long seed,
                                      unsigned long long offset,
                                      float* __restrict__ z3,
                                      float* __restrict__ a3,
                                      unsigned char* __restrict__ m3) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B * 128) return;
  int b = i / 128, j = i % 128;
  float t = b3[j];
  for (int s = 0; s < FC1B_SPLIT; ++s)
    t += slab[((long long)s * B + b) * 128 + j];
  z3[i] = t;
  float r = t > 0.f ? t : 0.f;
  unsigned char keep = 1;
  if (p2 > 0.f) {
    hiprandStatePhilox4_32_10_t st;
    hiprand_init(seed ^ 0x9e3779b97f4a7c15ull, (unsigned long long)i,
                 offset, &st);
    keep = hiprand_uniform(&st) >= p2;
  }
  m3[i] = keep;
  a3[i] = keep ? r / (1.f - p2) : 0.f;
}

// conv2 forward, bf16: im2col A tile [64][288] and W2 [64][288] (both
// k-contiguous bf16, stride 296) built in LDS per block; 36 bf16 MFMAs
// per wave replace 288 f32 MFMA issues.
#define C2_LD 296
__global__ __launch_bounds__(256)
void k_conv2_fwd_mfma_bf16(const float* __restrict__ a1,
                           const float* __restrict__ w2,
                           const float* __restrict__ b2, int B,
                           float* __restrict__ r2) {
  __shared__ __bf16 imc[64 * C2_LD];
  __shared__ __bf16 wb[64 * C2_LD];
  int b = blockIdx.x / 9, mt = blockIdx.x % 9;
  const float* a1b = a1 + (long long)b * 21632;
  for (int i = threadIdx.x; i < 64 * 288; i += 256) {
    int mr = i / 288, k = i % 288;
    int m = mt * 64 + mr, yy = m / 24, xx = m % 24;
    int ci = k / 9, rem = k % 9, kh = rem / 3, kw = rem % 3;
    imc[mr * C2_LD + k] = (__bf16)a1b[ci * 676 + (yy + kh) * 26 + xx + kw];
    wb[mr * C2_LD + k] = (__bf16)w2[i];  // mr doubles as co here
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  f32x4 acc[4] = {f32x4{0,0,0,0}, f32x4{0,0,0,0},
                  f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 288; k0 += 32) {
    bf16x8 a = ld_bf16x8(&imc[(w * 16 + il) * C2_LD + k0 + kc8]);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      bf16x8 bv = ld_bf16x8(&wb[(nt * 16 + il) * C2_LD + k0 + kc8]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc[nt], 0, 0, 0);
    }
  }
  int om = mt * 64 + w * 16 + (lane >> 4) * 4;
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      int co = nt * 16 + il;
      float v = acc[nt][r] + b2[co];
      r2[((long long)b * 64 + co) * 576 + om + r] = v > 0.f ? v : 0.f;
    }
}

// conv2 backward-data, bf16: padded-im2col A tile [64][576] of dz2 plus
// W2^T-per-ci tile [32][576] (k=(co,kh,kw) contiguous), stride 584.
#define C2B_LD 584
__global__ __launch_bounds__(256)
void k_conv2_bwd_x_mfma_bf16(const float* __restrict__ dz2,
                             const float* __restrict__ w2,
                             const float* __restrict__ a1, int B,
                             float* __restrict__ dz1) {
  __shared__ __bf16 imc[64 * C2B_LD];   // 74.8 KB
  __shared__ __bf16 wt[32 * C2B_LD];    // 37.4 KB
  int b = blockIdx.x / 11, mt = blockIdx.x % 11;
  const float* dzb = dz2 + (long long)b * 36864;
  for (int i = threadIdx.x; i < 64 * 576; i += 256) {
    int mr = i / 576, k = i % 576;
    int m = mt * 64 + mr, p = m / 26, q = m % 26;
    int co = k / 9, rem = k % 9, kh = rem / 3, kw = rem % 3;
    int y = p - kh, x = q - kw;
    imc[mr * C2B_LD + k] =
        (__bf16)((m < 676 && y >= 0 && y < 24 && x >= 0 && x < 24)
                     ? dzb[co * 576 + y * 24 + x] : 0.f);
  }
  for (int i = threadIdx.x; i < 32 * 576; i += 256) {
    int ci = i / 576, k = i % 576;
    int co = k / 9, rem = k % 9;
    wt[ci * C2B_LD + k] = (__bf16)w2[(long long)co * 288 + ci * 9 + rem];
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  f32x4 acc[2] = {f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 576; k0 += 32) {
    bf16x8 a = ld_bf16x8(&imc[(w * 16 + il) * C2B_LD + k0 + kc8]);
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      bf16x8 bv = ld_bf16x8(&wt[(nt * 16 + il) * C2B_LD + k0 + kc8]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc[nt], 0, 0, 0);
    }
  }
  int om = mt * 64 + w * 16 + (lane >> 4) * 4;
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    if (om + r >= 676) continue;
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      int ci = nt * 16 + il;
      long long o = ((long long)b * 32 + ci) * 676 + om + r;
      dz1[o] = a1[o] > 0.f ? acc[nt][r] : 0.f;
    }
  }
}

// conv2 backward-weight, bf16: dz2[b] as bf16 [64][576] (A, k=o
// contiguous) + im2col^T tile [48][576] of a1 (B); per-b slabs folded
// deterministically as in the f32 path.
__global__ __launch_bounds__(256)
void k_conv2_bwd_w_mfma_bf16(const float* __restrict__ dz2,
                             const float* __restrict__ a1, int B,
                             float* __restrict__ slab) {
  __shared__ __bf16 dzb16[64 * C2B_LD];  // 74.8 KB
  __shared__ __bf16 imt[48 * C2B_LD];    // 56.1 KB
  int nb = blockIdx.x % 6, b = blockIdx.x / 6;
  const float* dzb = dz2 + (long long)b * 36864;
  const float* a1b = a1 + (long long)b * 21632;
  for (int i = threadIdx.x; i < 64 * 576; i += 256) {
    int co = i / 576, o = i % 576;
    dzb16[co * C2B_LD + o] = (__bf16)dzb[i];
    (void)o;
  }
  for (int i = threadIdx.x; i < 48 * 576; i += 256) {
    int nr = i / 576, o = i % 576;
    int n = nb * 48 + nr;
    int ci = n / 9, rem = n % 9, kh = rem / 3, kw = rem % 3;
    int yy = o / 24, xx = o % 24;
    imt[nr * C2B_LD + o] = (__bf16)a1b[ci * 676 + (yy + kh) * 26 + xx + kw];
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  f32x4 acc[3] = {f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 576; k0 += 32) {
    bf16x8 a = ld_bf16x8(&dzb16[(w * 16 + il) * C2B_LD + k0 + kc8]);
    #pragma unroll
    for (int nt = 0; nt < 3; ++nt) {
      bf16x8 bv = ld_bf16x8(&imt[(nt * 16 + il) * C2B_LD + k0 + kc8]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc[nt], 0, 0, 0);
    }
  }
  float* out = slab + (long long)b * 18432;
  int orow = w * 16 + (lane >> 4) * 4;
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    #pragma unroll
    for (int nt = 0; nt < 3; ++nt)
      out[(orow + r) * 288 + nb * 48 + nt * 16 + il] = acc[nt][r];
}


// ---------------------------------------------------------------------------
// bf16 MFMA probe: one v_mfma_f32_16x16x32_bf16 with the documented
// fragment maps (A[i=l&15][k=(l>>4)*8+e], B[k=(l>>4)*8+e][j=l&15],
// D[row=(l>>4)*4+r][col=l&15]) — the numerics test pins the layout
// before the bf16 kernels build on it.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_;

__global__ void k_mfma_bf16_probe(const __bf16* __restrict__ A,   // [16][32]
                                  const __bf16* __restrict__ B,   // [32][16]
                                  float* __restrict__ D) {        // [16][16]
  int l = threadIdx.x;
  bf16x8 a, b;
  #pragma unroll
  for (int e = 0; e < 8; ++e) {
    int k = (l >> 4) * 8 + e;
    a[e] = A[(l & 15) * 32 + k];
    b[e] = B[k * 16 + (l & 15)];
  }
  f32x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
}

extern "C" void launch_fc1_fwd_mfma_bf16(const float* a2, const float* w3,
                              const float* b3, int B, float p2,
                              unsigned long long seed,
                              unsigned long long offset, float* slab,
                              float* z3, float* a3, unsigned char* m3,
                              hipStream_t s) {
  hipLaunchKernelGGL(k_fc1_fwd_mfma_bf16, dim3(FC1B_SPLIT), dim3(FBLK), 0, s,
                     a2, w3, B, slab);
  hipLaunchKernelGGL(k_fc1_fwd_reduce_bf16, dim3((B * 128 + FBLK - 1) / FBLK),
                     dim3(FBLK), 0, s, slab, b3, B, p2, seed, offset,
                     z3, a3, m3);
}
extern "C" void launch_conv2_fwd_mfma_bf16(const float* a1, const float* w2,
                                const float* b2, int B, float* r2,
                                hipStream_t s) {
  hipLaunchKernelGGL(k_conv2_fwd_mfma_bf16, dim3(B * 9), dim3(FBLK), 0, s,
                     a1, w2, b2, B, r2);
}
extern "C" void launch_conv2_bwd_x_mfma_bf16(const float* dz2, const float* w2,
                                  const float* a1, int B, float* dz1,
                                  hipStream_t s) {
  hipLaunchKernelGGL(k_conv2_bwd_x_mfma_bf16, dim3(B * 11), dim3(FBLK), 0, s,
                     dz2, w2, a1, B, dz1);
}
extern "C" void launch_conv2_bwd_w_mfma_bf16(const float* dz2, const float* a1, int B,
                                  float* slab, float* dw2, float* db2,
                                  hipStream_t s) {
  hipLaunchKernelGGL(k_conv2_bwd_w_mfma_bf16, dim3(6 * B), dim3(FBLK), 0, s,
                     dz2, a1, B, slab);
  hipLaunchKernelGGL(k_conv2_bwd_w_fold, dim3((18432 + FBLK - 1) / FBLK),
                     dim3(FBLK), 0, s, slab, B, dw2);
  hipLaunchKernelGGL(k_conv2_bwd_b, dim3(64), dim3(FBLK), 0, s, dz2, B, db2);
}
extern "C" void launch_mfma_bf16_probe(const void* A, const void* B, float* D,
                                       hipStream_t s) {
  hipLaunchKernelGGL(k_mfma_bf16_probe, dim3(1), dim3(64), 0, s,
                     (const __bf16*)A, (const __bf16*)B, D);
}


// CNN_EPOCH_DEBUG=1: synchronize + check after every launch so a device
// fault names the kernel that raised it (diagnostics only — the hot path
// never syncs)
#include <cstdio>
#include <cstdlib>
#define EPOCH_CHK(name)                                                    \
  do {                                                                     \
    if (epoch_dbg) {                                                       \
      hipError_t e_ = hipStreamSynchronize(s);                             \
      hipError_t e2_ = hipGetLastError();                                  \
      if (e_ != hipSuccess || e2_ != hipSuccess) {                         \
        fprintf(stderr, "cnn_epoch fault after %s (it=%d B=%d): %s / %s\n",\
                name, it, B, hipGetErrorString(e_), hipGetErrorString(e2_));\
        fflush(stderr);                                                    \
        abort();                                                           \
      }                                                                    \
    }                                                                      \
  } while (0)

extern "C" void launch_cnn_epoch(
    const float* shard_x, const long long* shard_y, const long long* order,
    long long n, int bs, int C, float* params, float* grads,
    CnnWorkspace ws, const float* lr_t, float max_norm, float p1, float p2,
    float* stats_acc, float* loss_acc, unsigned long long seed,
    hipStream_t s, long long row_base = 0, int use_bf16 = 0) {
  static const bool epoch_dbg = getenv("CNN_EPOCH_DEBUG") != nullptr;
  CnnOffsets o = cnn_offsets(C);
  int n_batches = (int)((n + bs - 1) / bs);
  for (int it = 0; it < n_batches; ++it) {
    long long start = (long long)it * bs;
    int B = (int)((start + bs <= n) ? bs : (n - start));
    unsigned long long off = (unsigned long long)it;
    int g1 = (B * 784 + FBLK - 1) / FBLK;
    hipLaunchKernelGGL(k_gather_batch, dim3(g1), dim3(FBLK), 0, s,
                       shard_x, shard_y, order, start, row_base, B,
                       ws.xb, ws.yb);
    EPOCH_CHK("k_gather_batch");
    hipLaunchKernelGGL(k_conv1_fwd, dim3((B * 21632 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, ws.xb, params + o.w1, params + o.b1,
                       B, ws.a1);
    EPOCH_CHK("k_conv1_fwd");
    if (use_bf16) {
      // bf16 kernels read w2 directly (LDS staging converts)
      hipLaunchKernelGGL(k_conv2_fwd_mfma_bf16, dim3(B * 9), dim3(FBLK),
                         0, s, ws.a1, params + o.w2, params + o.b2, B, ws.r2);
    } else {
      hipLaunchKernelGGL(k_w2_layouts, dim3((18432 + FBLK - 1) / FBLK),
                         dim3(FBLK), 0, s, params + o.w2, ws.w2t, ws.w2rot);
      EPOCH_CHK("k_w2_layouts");
      hipLaunchKernelGGL(k_conv2_fwd_mfma, dim3(B * 9), dim3(FBLK),
                         0, s, ws.a1, ws.w2t, params + o.b2, B, ws.r2);
    }
    EPOCH_CHK("k_conv2_fwd_mfma");
    hipLaunchKernelGGL(k_pool_drop_fwd, dim3((B * 9216 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, ws.r2, B, p1, seed, off, ws.a2,
                       ws.pidx, ws.m2);
    EPOCH_CHK("k_pool_drop_fwd");
    if (use_bf16) {
      hipLaunchKernelGGL(k_fc1_fwd_mfma_bf16, dim3(FC1B_SPLIT), dim3(FBLK),
                         0, s, ws.a2, params + o.w3, B, ws.wsl);
      EPOCH_CHK("k_fc1_fwd_mfma_bf16");
      hipLaunchKernelGGL(k_fc1_fwd_reduce_bf16,
                         dim3((B * 128 + FBLK - 1) / FBLK),
                         dim3(FBLK), 0, s, ws.wsl, params + o.b3, B, p2, seed,
                         off, ws.z3, ws.a3, ws.m3);
    } else {
      hipLaunchKernelGGL(k_fc1_fwd_mfma, dim3(FC1_SPLIT), dim3(FBLK), 0, s,
                         ws.a2, params + o.w3, B, ws.wsl);
      EPOCH_CHK("k_fc1_fwd_mfma");
      hipLaunchKernelGGL(k_fc1_fwd_reduce, dim3((B * 128 + FBLK - 1) / FBLK),
                         dim3(FBLK), 0, s, ws.wsl, params + o.b3, B, p2, seed,
                         off, ws.z3, ws.a3, ws.m3);
    }
    EPOCH_CHK("k_fc1_fwd_reduce");
    hipLaunchKernelGGL(k_fc2_loss_fwd, dim3(B), dim3(FBLK),
                       C * (int)sizeof(float), s, ws.a3, params + o.w4,
                       params + o.b4, ws.yb, B, C, ws.dlogits, loss_acc);
    EPOCH_CHK("k_fc2_loss_fwd");
    hipLaunchKernelGGL(k_fc2_bwd_w, dim3((C * 128 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, ws.dlogits, ws.a3, B, C,
                       grads + o.w4, grads + o.b4);
    EPOCH_CHK("k_fc2_bwd_w");
    hipLaunchKernelGGL(k_fc2_bwd_x, dim3((B * 128 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, ws.dlogits, params + o.w4, ws.z3,
                       ws.m3, B, C, p2, ws.dz3);
    EPOCH_CHK("k_fc2_bwd_x");
    hipLaunchKernelGGL(k_fc1_bwd_w_mfma, dim3(144), dim3(FBLK), 0, s,
                       ws.dz3, ws.a2, B, grads + o.w3);
    EPOCH_CHK("k_fc1_bwd_w_mfma");
    hipLaunchKernelGGL(k_fc1_bwd_b, dim3(1), dim3(128), 0, s,
                       ws.dz3, B, grads + o.b3);
    EPOCH_CHK("k_fc1_bwd_b");
    hipLaunchKernelGGL(k_fc1_bwd_x_mfma, dim3(144), dim3(FBLK), 0, s,
                       ws.dz3, params + o.w3, B, ws.da2);
    EPOCH_CHK("k_fc1_bwd_x_mfma");
    hipLaunchKernelGGL(k_pool_drop_bwd, dim3((B * 9216 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, ws.da2, ws.pidx, ws.m2, ws.r2, B,
                       p1, ws.dz2);
    EPOCH_CHK("k_pool_drop_bwd");
    if (use_bf16) {
      hipLaunchKernelGGL(k_conv2_bwd_w_mfma_bf16, dim3(6 * B), dim3(FBLK),
                         0, s, ws.dz2, ws.a1, B, ws.wsl);
    } else {
      hipLaunchKernelGGL(k_dz2_transpose, dim3((B * 36864 + FBLK - 1) / FBLK),
                         dim3(FBLK), 0, s, ws.dz2, B, ws.r2);
      EPOCH_CHK("k_dz2_transpose");
      hipLaunchKernelGGL(k_conv2_bwd_w_mfma, dim3(6 * B), dim3(FBLK), 0, s,
                         ws.r2, ws.a1, B, ws.wsl);
    }
    EPOCH_CHK("k_conv2_bwd_w_mfma");
    hipLaunchKernelGGL(k_conv2_bwd_w_fold, dim3((18432 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, ws.wsl, B, grads + o.w2);
    EPOCH_CHK("k_conv2_bwd_w_fold");
    hipLaunchKernelGGL(k_conv2_bwd_b, dim3(64), dim3(FBLK), 0, s,
                       ws.dz2, B, grads + o.b2);
    EPOCH_CHK("k_conv2_bwd_b");
    if (use_bf16)
      hipLaunchKernelGGL(k_conv2_bwd_x_mfma_bf16, dim3(B * 11), dim3(FBLK),
                         0, s, ws.dz2, params + o.w2, ws.a1, B, ws.dz1);
    else
      hipLaunchKernelGGL(k_conv2_bwd_x_mfma, dim3(B * 11), dim3(FBLK),
                         0, s, ws.dz2, ws.w2rot, ws.a1, B, ws.dz1);
    EPOCH_CHK("k_conv2_bwd_x_mfma");
    hipLaunchKernelGGL(k_conv1_bwd_w, dim3(32), dim3(1024), 0, s,
                       ws.xb, ws.dz1, B, grads + o.w1, grads + o.b1);
    EPOCH_CHK("k_conv1_bwd_w");
    // fused clip + sufficient stats + SGD on the whole arena
    hipMemsetAsync(ws.red_acc, 0, 2 * sizeof(double), s);
    launch_sum_sumsq2(grads, o.total, ws.red_partials, ws.red_acc, s);
    EPOCH_CHK("launch_sum_sumsq2");
    launch_clip_apply_stats(grads, o.total, ws.red_acc, max_norm, 1e-6f,
                            stats_acc, s);
    EPOCH_CHK("launch_clip_apply_stats");
    launch_sgd_step(params, grads, nullptr, 0.f, lr_t, 0.f, 0.f, 0.f, 0, 0,
                    o.total, s);
    EPOCH_CHK("launch_sgd_step");
  }
}

// ---------------------------------------------------------------------------
// WHOLE-ROUND driver: trains K clients back-to-back in one host call —
// per client: copy-in server weights, run the fused epoch, write the
// weighted pseudo-gradient, accumulate into the round buffer.  Per-client
// losses/stats land in slots of loss_out[K] / stats_out[K*2].
// ---------------------------------------------------------------------------
extern "C" void launch_cnn_round(
    const float* shard_x, const long long* shard_y,
    const long long* orders,            // concatenated per-client shuffles
    const long long* row_bases,         // HOST: K shard row offsets
    const long long* order_offs,        // HOST: K offsets into `orders`
    const long long* counts,            // HOST: K sample counts
    const float* weights,               // HOST: K aggregation weights
    const unsigned long long* seeds,    // HOST: K dropout seeds
    int K, int bs, int C,
    const float* server_params, float* params, float* grads,
    float* round_accum, CnnWorkspace ws, const float* lr_t, float max_norm,
    float p1, float p2, float* stats_out, float* loss_out,
    hipStream_t s, int use_bf16) {
  CnnOffsets o = cnn_offsets(C);
  int gp = (int)((o.total + FBLK - 1) / FBLK);
  if (gp > 2048) gp = 2048;
  for (int k = 0; k < K; ++k) {
    hipLaunchKernelGGL(k_copy, dim3(gp), dim3(FBLK), 0, s,
                       params, server_params, o.total);
    launch_cnn_epoch(shard_x, shard_y, orders + order_offs[k], counts[k],
                     bs, C, params, grads, ws, lr_t, max_norm, p1, p2,
                     stats_out + 2 * k, loss_out + k, seeds[k], s,
                     row_bases[k], use_bf16);
    hipLaunchKernelGGL(k_cnn_pseudo_grad, dim3(gp), dim3(FBLK), 0, s,
                       grads, server_params, params, weights[k], o.total);
    hipLaunchKernelGGL(k_cnn_axpy, dim3(gp), dim3(FBLK), 0, s,
                       round_accum, grads, o.total);
  }
}

// ---------------------------------------------------------------------------
// debug/test launchers for the MFMA kernels (numerics tests call these
// one-by-one against torch references — tests/test_mfma_gpu.py)
// ---------------------------------------------------------------------------
extern "C" {
void launch_w2_layouts(const float* w2, float* w2t, float* w2rot,
                       hipStream_t s) {
  hipLaunchKernelGGL(k_w2_layouts, dim3((18432 + FBLK - 1) / FBLK),
                     dim3(FBLK), 0, s, w2, w2t, w2rot);
}
void launch_conv2_fwd_mfma(const float* a1, const float* w2t, const float* b2,
                           int B, float* r2, hipStream_t s) {
  hipLaunchKernelGGL(k_conv2_fwd_mfma, dim3(B * 9), dim3(FBLK), 0, s,
                     a1, w2t, b2, B, r2);
}
void launch_conv2_bwd_x_mfma(const float* dz2, const float* w2rot,
                             const float* a1, int B, float* dz1,
                             hipStream_t s) {
  hipLaunchKernelGGL(k_conv2_bwd_x_mfma, dim3(B * 11), dim3(FBLK), 0, s,
                     dz2, w2rot, a1, B, dz1);
}
void launch_conv2_bwd_w_mfma(const float* dz2, const float* a1, int B,
                             float* dz2t, float* slab, float* dw2, float* db2,
                             hipStream_t s) {
  hipLaunchKernelGGL(k_dz2_transpose, dim3((B * 36864 + FBLK - 1) / FBLK),
                     dim3(FBLK), 0, s, dz2, B, dz2t);
  hipLaunchKernelGGL(k_conv2_bwd_w_mfma, dim3(6 * B), dim3(FBLK), 0, s,
                     dz2t, a1, B, slab);
  hipLaunchKernelGGL(k_conv2_bwd_w_fold, dim3((18432 + FBLK - 1) / FBLK),
                     dim3(FBLK), 0, s, slab, B, dw2);
  hipLaunchKernelGGL(k_conv2_bwd_b, dim3(64), dim3(FBLK), 0, s, dz2, B, db2);
}
void launch_fc1_fwd_mfma(const float* a2, const float* w3, const float* b3,
                         int B, float p2, unsigned long long seed,
                         unsigned long long offset, float* slab, float* z3,
                         float* a3, unsigned char* m3, hipStream_t s) {
  hipLaunchKernelGGL(k_fc1_fwd_mfma, dim3(FC1_SPLIT), dim3(FBLK), 0, s,
                     a2, w3, B, slab);
  hipLaunchKernelGGL(k_fc1_fwd_reduce, dim3((B * 128 + FBLK - 1) / FBLK),
                     dim3(FBLK), 0, s, slab, b3, B, p2, seed, offset,
                     z3, a3, m3);
}
void launch_fc1_bwd_w_mfma(const float* dz3, const float* a2, int B,
                           float* dw3, float* db3, hipStream_t s) {
  hipLaunchKernelGGL(k_fc1_bwd_w_mfma, dim3(144), dim3(FBLK), 0, s,
                     dz3, a2, B, dw3);
  hipLaunchKernelGGL(k_fc1_bwd_b, dim3(1), dim3(128), 0, s, dz3, B, db3);
}
void launch_fc1_bwd_x_mfma(const float* dz3, const float* w3, int B,
                           float* da2, hipStream_t s) {
  hipLaunchKernelGGL(k_fc1_bwd_x_mfma, dim3(144), dim3(FBLK), 0, s,
                     dz3, w3, B, da2);
}
}
