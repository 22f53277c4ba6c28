// Cross-client MEGA-batched CNN round for gfx950.
//
// The per-client fused round (fused_cnn.hip launch_cnn_round) enqueues
// each client's epoch on its own stream; concurrent small kernels only
// co-schedule ~2.5x on this stack (tools/diag_streams.py), and per-client
// grids (e.g. conv2 fwd: B*9 = 180 blocks) underfill the 256-CU chip.
// Here ONE launch set per batch-step trains ALL K sampled clients: the
// super-batch dimension is G = K*bs rows (row g belongs to client
// k = g/bs), every kernel takes per-client parameter/gradient stacks
// (params_stack[k*P ..]), and grids scale by K — conv2 fwd becomes
// K*bs*9 = 1800 blocks at the benchmark shape.
//
// Ragged handling is free of masks in the backward chain: the gather
// zero-fills inactive rows (b >= B_k(t)) and writes yb = -1, the loss
// kernel zeroes those rows' dlogits, and every weight gradient is then
// exactly the per-client value (zero rows contribute nothing).  Clients
// whose epochs ended (t >= ceil(count/bs)) naturally produce zero
// gradients and their SGD step no-ops.  Dropout Philox streams use
// CLIENT-LOCAL indices, so masks are bit-identical to the per-client
// fused path for the same (seed, batch) keys.

#include <hip/hip_runtime.h>
#include <hiprand/hiprand_kernel.h>

#define FBLK 256

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define MC2F_LD 585  /* padded f32 LDS row stride (585 %% 32 = 9, odd ->
                        conflict-free 16-lane fragment groups) */

struct MegaOffsets {
  long long w1, b1, w2, b2, w3, b3, w4, b4, total;
};

static MegaOffsets mega_offsets(int C) {
  MegaOffsets o;
  o.w1 = 0; o.b1 = 288; o.w2 = 320; o.b2 = 18752; o.w3 = 18816;
  o.b3 = o.w3 + 1179648; o.w4 = o.b3 + 128;
  o.b4 = o.w4 + (long long)C * 128; o.total = o.b4 + C;
  return o;
}

// per-row helpers: g in [0, K*bs), k = g / bs, b = g % bs
__device__ __forceinline__ int mega_Bk(const long long* counts, int k,
                                       int t, int bs) {
  long long rem = counts[k] - (long long)t * bs;
  return rem <= 0 ? 0 : (rem < bs ? (int)rem : bs);
}

__global__ void k_copy_stack(float* __restrict__ dst,
                             const float* __restrict__ src, long long P,
                             int K) {
  long long total = (long long)K * P;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x)
    dst[i] = src[i % P];
}

__global__ void k_gather_mb(const float* __restrict__ shard_x,
                            const long long* __restrict__ shard_y,
                            const long long* __restrict__ orders,
                            const long long* __restrict__ row_bases,
                            const long long* __restrict__ order_offs,
                            const long long* __restrict__ counts,
                            int t, int bs, int K,
                            float* __restrict__ xb, int* __restrict__ yb) {
  int G = K * bs;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < (long long)G * 784; i += (long long)gridDim.x * blockDim.x) {
    int g = (int)(i / 784), j = (int)(i % 784);
    int k = g / bs, b = g % bs;
    int Bk = mega_Bk(counts, k, t, bs);
    if (b < Bk) {
      long long src = row_bases[k]
          + orders[order_offs[k] + (long long)t * bs + b];
      xb[i] = shard_x[src * 784 + j];
      if (j == 0) yb[g] = (int)shard_y[src];
    } else {
      xb[i] = 0.f;
      if (j == 0) yb[g] = -1;
    }
  }
}

__global__ void k_conv1_fwd_mb(const float* __restrict__ x,
                               const float* __restrict__ params, long long P,
                               long long ow1, long long ob1, int bs, int K,
                               float* __restrict__ a1) {
  long long total = (long long)K * bs * 21632;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int xx = (int)(i % 26), yy = (int)((i / 26) % 26);
    int co = (int)((i / 676) % 32);
    long long g = i / 21632;
    int k = (int)(g / bs);
    const float* w1 = params + (long long)k * P + ow1;
    const float* b1 = params + (long long)k * P + ob1;
    const float* xp = x + g * 784 + yy * 28 + xx;
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

__global__ void k_w2_layouts_mb(const float* __restrict__ params,
                                long long P, long long ow2, int K,
                                float* __restrict__ w2t_stack,
                                float* __restrict__ w2rot_stack) {
  long long total = (long long)K * 18432;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i / 18432), r = (int)(i % 18432);
    int co = r / 288, kk = r % 288;
    float v = params[(long long)k * P + ow2 + r];
    w2t_stack[(long long)k * 18432 + kk * 64 + co] = v;
    int rem = kk % 9;
    w2rot_stack[(long long)k * 18432 + ((long long)co * 9 + rem) * 32
                + kk / 9] = v;
  }
}

// conv2 forward (f32 MFMA): grid = G * 9 blocks; per-block client k
__global__ __launch_bounds__(256)
void k_conv2_fwd_mfma_mb(const float* __restrict__ a1,
                         const float* __restrict__ w2t_stack,
                         const float* __restrict__ params, long long P,
                         long long ob2, int bs, int K,
                         float* __restrict__ r2) {
  __shared__ float lds[32 * 676];
  long long g = blockIdx.x / 9;
  int mt = blockIdx.x % 9;
  int k = (int)(g / bs);
  const float* src = a1 + g * 21632;
  for (int i = threadIdx.x; i < 21632; i += 256) lds[i] = src[i];
  __syncthreads();
  const float* w2t = w2t_stack + (long long)k * 18432;
  const float* b2 = params + (long long)k * P + ob2;
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int m = mt * 64 + w * 16 + (lane & 15);
  int yy = m / 24, xx = m % 24;
  int kc = lane >> 4;
  f32x4 acc[4] = {f32x4{0,0,0,0}, f32x4{0,0,0,0},
                  f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  #pragma unroll 4
  for (int k0 = 0; k0 < 288; k0 += 4) {
    int kk = k0 + kc;
    int ci = kk / 9, rem = kk % 9, kh = rem / 3, kw = rem % 3;
    float a = lds[ci * 676 + (yy + kh) * 26 + xx + kw];
    const float* wrow = w2t + (long long)kk * 64 + (lane & 15);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      float bv = wrow[nt * 16];
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
      r2[(g * 64 + co) * 576 + om + r] = v > 0.f ? v : 0.f;
    }
}

__global__ void k_pool_drop_fwd_mb(const float* __restrict__ r2, int bs,
                                   int K, float p1,
                                   const long long* __restrict__ seeds,
                                   unsigned long long offset,
                                   float* __restrict__ a2,
                                   unsigned char* __restrict__ pidx,
                                   unsigned char* __restrict__ m2) {
  long long total = (long long)K * bs * 9216;
  float inv_keep = (p1 < 1.f) ? 1.f / (1.f - p1) : 0.f;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int px = (int)(i % 12), py = (int)((i / 12) % 12);
    int c = (int)((i / 144) % 64);
    long long g = i / 9216;
    int k = (int)(g / bs), b = (int)(g % bs);
    const float* base = r2 + ((g * 64 + c) * 24 + 2 * py) * 24 + 2 * px;
    float v0 = base[0], v1 = base[1], v2 = base[24], v3 = base[25];
    float m = v0; int idx = 0;
    if (v1 > m) { m = v1; idx = 1; }
    if (v2 > m) { m = v2; idx = 2; }
    if (v3 > m) { m = v3; idx = 3; }
    unsigned char keep = 1;
    if (p1 > 0.f) {
      // CLIENT-LOCAL element index -> bitwise-identical masks to the
      // per-client fused path
      long long li = (long long)b * 9216 + (i % 9216);
      hiprandStatePhilox4_32_10_t st;
      hiprand_init((unsigned long long)seeds[k],
                   (unsigned long long)(li >> 2), offset, &st);
      float4 u = hiprand_uniform4(&st);
      float uu = (li & 3) == 0 ? u.x : (li & 3) == 1 ? u.y
                 : (li & 3) == 2 ? u.z : u.w;
      keep = uu >= p1;
    }
    pidx[i] = (unsigned char)idx;
    m2[i] = keep;
    a2[i] = keep ? m * inv_keep : 0.f;
  }
}

// fc1 forward (f32 MFMA, LDS-staged): grid = K * FC1_SPLIT blocks
#define FC1M_SPLIT 64
#define FC1M_CH (9216 / FC1M_SPLIT)
#define FC1M_LD (FC1M_CH + 1)
__global__ __launch_bounds__(256)
void k_fc1_fwd_mfma_mb(const float* __restrict__ a2,
                       const float* __restrict__ params, long long P,
                       long long ow3, int bs, int K,
                       float* __restrict__ slab) {
  __shared__ float lw[128 * FC1M_LD];
  __shared__ float la[32 * FC1M_LD];
  int k = blockIdx.x / FC1M_SPLIT;
  int s = blockIdx.x % FC1M_SPLIT;
  int k_base = s * FC1M_CH;
  const float* w3 = params + (long long)k * P + ow3;
  const float* a2k = a2 + (long long)k * bs * 9216;
  for (int i = threadIdx.x; i < 128 * FC1M_CH; i += 256) {
    int row = i / FC1M_CH, kk = i % FC1M_CH;
    lw[row * FC1M_LD + kk] = w3[(long long)row * 9216 + k_base + kk];
  }
  for (int i = threadIdx.x; i < 32 * FC1M_CH; i += 256) {
    int bu = i / FC1M_CH, kk = i % FC1M_CH;
    la[bu * FC1M_LD + kk] = bu < bs
        ? a2k[(long long)bu * 9216 + k_base + kk] : 0.f;
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int kc = lane >> 4, il = lane & 15;
  f32x4 acc[2][2] = {{f32x4{0,0,0,0}, f32x4{0,0,0,0}},
                     {f32x4{0,0,0,0}, f32x4{0,0,0,0}}};
  for (int k0 = 0; k0 < FC1M_CH; k0 += 4) {
    int kk = k0 + kc;
    float a0 = lw[(w * 32 + il) * FC1M_LD + kk];
    float a1v = lw[(w * 32 + 16 + il) * FC1M_LD + kk];
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      float bv = la[(u * 16 + il) * FC1M_LD + kk];
      acc[0][u] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, bv, acc[0][u], 0, 0, 0);
      acc[1][u] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1v, bv, acc[1][u], 0, 0, 0);
    }
  }
  float* slk = slab + (long long)k * FC1M_SPLIT * bs * 128;
  #pragma unroll
  for (int tt = 0; tt < 2; ++tt)
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      int bu = u * 16 + il;
      if (bu >= bs) continue;
      int j = w * 32 + tt * 16 + (lane >> 4) * 4;
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        slk[((long long)s * bs + bu) * 128 + j + r] = acc[tt][u][r];
    }
}

__global__ void k_fc1_fwd_reduce_mb(const float* __restrict__ slab,
                                    const float* __restrict__ params,
                                    long long P, long long ob3, int bs,
                                    int K, float p2,
                                    const long long* __restrict__ seeds,
                                    unsigned long long offset,
                                    float* __restrict__ z3,
                                    float* __restrict__ a3,
                                    unsigned char* __restrict__ m3) {
  long long total = (long long)K * bs * 128;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    long long g = i / 128;
    int j = (int)(i % 128);
    int k = (int)(g / bs), b = (int)(g % bs);
    const float* slk = slab + (long long)k * FC1M_SPLIT * bs * 128;
    float t = params[(long long)k * P + ob3 + j];
    for (int s = 0; s < FC1M_SPLIT; ++s)
      t += slk[((long long)s * bs + b) * 128 + j];
    z3[i] = t;
    float r = t > 0.f ? t : 0.f;
    unsigned char keep = 1;
    if (p2 > 0.f) {
      long long li = (long long)b * 128 + j;  // client-local
      hiprandStatePhilox4_32_10_t st;
      hiprand_init((unsigned long long)seeds[k] ^ 0x9e3779b97f4a7c15ull,
                   (unsigned long long)li, offset, &st);
      keep = hiprand_uniform(&st) >= p2;
    }
    m3[i] = keep;
    a3[i] = keep ? r / (1.f - p2) : 0.f;
  }
}

// fc2 + softmax + CE: one block per super-row g; yb<0 rows zero dlogits
__global__ void k_fc2_loss_fwd_mb(const float* __restrict__ a3,
                                  const float* __restrict__ params,
                                  long long P, long long ow4, long long ob4,
                                  const long long* __restrict__ counts,
                                  int t, int bs, int K, int C,
                                  const int* __restrict__ yb,
                                  float* __restrict__ dlogits,
                                  float* __restrict__ loss_out) {
  extern __shared__ float sm[];
  long long g = blockIdx.x;
  int k = (int)(g / bs);
  int target = yb[g];
  if (target < 0) {
    for (int j = threadIdx.x; j < C; j += blockDim.x)
      dlogits[g * C + j] = 0.f;
    return;
  }
  int Bk = mega_Bk(counts, k, t, bs);
  const float* ap = a3 + g * 128;
  const float* w4 = params + (long long)k * P + ow4;
  const float* b4 = params + (long long)k * P + ob4;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    const float* wp = w4 + (long long)j * 128;
    float s = b4[j];
    #pragma unroll 4
    for (int kk = 0; kk < 128; ++kk) s = fmaf(wp[kk], ap[kk], s);
    sm[j] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float mx = sm[0];
    for (int j = 1; j < C; ++j) mx = fmaxf(mx, sm[j]);
    float z = 0.f;
    for (int j = 0; j < C; ++j) { sm[j] = __expf(sm[j] - mx); z += sm[j]; }
    float inv = 1.f / z;
    for (int j = 0; j < C; ++j) {
      float p = sm[j] * inv;
      dlogits[g * C + j] = (p - (j == target ? 1.f : 0.f)) / (float)Bk;
    }
    atomicAdd(loss_out + k,
              -__logf(fmaxf(sm[target] * inv, 1e-30f)) / (float)Bk);
  }
}

// fc2 backward (weights): grid = K * ceil(C*128/256)
__global__ void k_fc2_bwd_w_mb(const float* __restrict__ dlogits,
                               const float* __restrict__ a3,
                               float* __restrict__ grads, long long P,
                               long long ow4, long long ob4, int bs, int K,
                               int C) {
  int per_k = (C * 128 + FBLK - 1) / FBLK;
  int k = blockIdx.x / per_k;
  int i0 = (blockIdx.x % per_k) * FBLK + threadIdx.x;
  if (i0 >= C * 128) return;
  int kk = i0 % 128, j = i0 / 128;
  const float* dlk = dlogits + (long long)k * bs * C;
  const float* a3k = a3 + (long long)k * bs * 128;
  float s = 0.f, sb = 0.f;
  for (int b = 0; b < bs; ++b) {
    float d = dlk[(long long)b * C + j];
    s = fmaf(d, a3k[(long long)b * 128 + kk], s);
    if (kk == 0) sb += d;
  }
  grads[(long long)k * P + ow4 + i0] = s;
  if (kk == 0) grads[(long long)k * P + ob4 + j] = sb;
}

__global__ void k_fc2_bwd_x_mb(const float* __restrict__ dlogits,
                               const float* __restrict__ params, long long P,
                               long long ow4, const float* __restrict__ z3,
                               const unsigned char* __restrict__ m3, int bs,
                               int K, int C, float p2,
                               float* __restrict__ dz3) {
  long long total = (long long)K * bs * 128;
  float inv_keep = 1.f / (1.f - p2);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int kk = (int)(i % 128);
    long long g = i / 128;
    int k = (int)(g / bs);
    const float* w4 = params + (long long)k * P + ow4;
    const float* dl = dlogits + g * C;
    float s = 0.f;
    for (int j = 0; j < C; ++j)
      s = fmaf(dl[j], w4[(long long)j * 128 + kk], s);
    float gg = (p2 > 0.f) ? (m3[i] ? s * inv_keep : 0.f) : s;
    dz3[i] = z3[i] > 0.f ? gg : 0.f;
  }
}

// fc1 backward weights (f32 MFMA): grid = K * 144
__global__ __launch_bounds__(256)
void k_fc1_bwd_w_mfma_mb(const float* __restrict__ dz3,
                         const float* __restrict__ a2,
                         float* __restrict__ grads, long long P,
                         long long ow3, int bs, int K) {
  int k = blockIdx.x / 144;
  int nblk = blockIdx.x % 144;
  const float* dz = dz3 + (long long)k * bs * 128;
  const float* a2k = a2 + (long long)k * bs * 9216;
  float* dw3 = grads + (long long)k * P + ow3;
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int kc = lane >> 4, il = lane & 15;
  f32x4 acc[2][4] = {{f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}},
                     {f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}, f32x4{0,0,0,0}}};
  for (int k0 = 0; k0 < ((bs + 3) & ~3); k0 += 4) {
    int b = k0 + kc;
    bool kv = b < bs;
    float a0 = kv ? dz[(long long)b * 128 + w * 32 + il] : 0.f;
    float a1v = kv ? dz[(long long)b * 128 + w * 32 + 16 + il] : 0.f;
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      long long n = (long long)nblk * 64 + nt * 16 + il;
      float bv = kv ? a2k[(long long)b * 9216 + n] : 0.f;
      acc[0][nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, bv, acc[0][nt], 0, 0, 0);
      acc[1][nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1v, bv, acc[1][nt], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int tt = 0; tt < 2; ++tt) {
    int j = w * 32 + tt * 16 + (lane >> 4) * 4;
    #pragma unroll
    for (int r = 0; r < 4; ++r)
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        dw3[(long long)(j + r) * 9216 + nblk * 64 + nt * 16 + il] =
            acc[tt][nt][r];
  }
}

__global__ void k_fc1_bwd_b_mb(const float* __restrict__ dz3,
                               float* __restrict__ grads, long long P,
                               long long ob3, int bs, int K) {
  int k = blockIdx.x;
  int j = threadIdx.x;
  if (j >= 128) return;
  const float* dz = dz3 + (long long)k * bs * 128;
  float s = 0.f;
  for (int b = 0; b < bs; ++b) s += dz[(long long)b * 128 + j];
  grads[(long long)k * P + ob3 + j] = s;
}

// fc1 backward data (f32 MFMA): grid = K * 144
__global__ __launch_bounds__(256)
void k_fc1_bwd_x_mfma_mb(const float* __restrict__ dz3,
                         const float* __restrict__ params, long long P,
                         long long ow3, int bs, int K,
                         float* __restrict__ da2) {
  int k = blockIdx.x / 144;
  int nblk = blockIdx.x % 144;
  const float* dz = dz3 + (long long)k * bs * 128;
  const float* w3 = params + (long long)k * P + ow3;
  float* da = da2 + (long long)k * bs * 9216;
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int kc = lane >> 4, il = lane & 15;
  f32x4 acc[2] = {f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 128; k0 += 4) {
    int j = k0 + kc;
    long long n = (long long)nblk * 64 + w * 16 + il;
    float bv = w3[(long long)j * 9216 + n];
    #pragma unroll
    for (int tt = 0; tt < 2; ++tt) {
      int bu = tt * 16 + il;
      float a = bu < bs ? dz[(long long)bu * 128 + j] : 0.f;
      acc[tt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[tt], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int tt = 0; tt < 2; ++tt) {
    int bu = tt * 16 + (lane >> 4) * 4;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (bu + r >= bs) continue;
      da[(long long)(bu + r) * 9216 + (long long)nblk * 64 + w * 16 + il] =
          acc[tt][r];
    }
  }
}

__global__ void k_pool_drop_bwd_mb(const float* __restrict__ da2,
                                   const unsigned char* __restrict__ pidx,
                                   const unsigned char* __restrict__ m2,
                                   const float* __restrict__ r2, int bs,
                                   int K, float p1,
                                   float* __restrict__ dz2) {
  long long total = (long long)K * bs * 9216;
  float inv_keep = (p1 < 1.f) ? 1.f / (1.f - p1) : 0.f;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int px = (int)(i % 12), py = (int)((i / 12) % 12);
    int c = (int)((i / 144) % 64);
    long long g = i / 9216;
    float gg = (p1 > 0.f) ? (m2[i] ? da2[i] * inv_keep : 0.f) : da2[i];
    int idx = pidx[i];
    long long base = ((g * 64 + c) * 24 + 2 * py) * 24 + 2 * px;
    #pragma unroll
    for (int d = 0; d < 4; ++d) {
      long long o = base + (d >> 1) * 24 + (d & 1);
      dz2[o] = (d == idx && r2[o] > 0.f) ? gg : 0.f;
    }
  }
}

__global__ void k_dz2_transpose_mb(const float* __restrict__ dz2, int bs,
                                   int K, float* __restrict__ dz2t) {
  long long total = (long long)K * bs * 36864;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    long long g = i / 36864;
    int r = (int)(i % 36864), o = r / 64, co = r % 64;
    dz2t[i] = dz2[(g * 64 + co) * 576 + o];
  }
}

// conv2 backward weights (f32 MFMA): 32x32 (co x n) tiles, K=576
// staged in TWO 288-deep chunks so the live LDS is 74 KB -> 2
// blocks/CU (a one-shot 576-deep stage was 149 KB -> 1 block/CU and
// measured WORSE than the 64-row form).  A = dz2t rows; B = im2col^T.
// grid = G * 18 (2 co-halves x 9 n-blocks); one accumulator per wave.
#define MC2FW_LD 289  /* 289 %% 32 = 1 -> conflict-free fragment groups */
__global__ __launch_bounds__(256)
void k_conv2_bwd_w_mfma_mb(const float* __restrict__ dz2t,
                           const float* __restrict__ a1, int bs, int K,
                           float* __restrict__ slab) {
  __shared__ float dzl[32 * MC2FW_LD];
  __shared__ float imt[32 * MC2FW_LD];
  int blk = blockIdx.x % 18;
  int ch = blk / 9, nb = blk % 9;
  long long g = blockIdx.x / 18;
  const float* dzb = dz2t + g * 36864;   // [o][co] layout
  const float* a1b = a1 + g * 21632;
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc = lane >> 4;
  int csub = w & 1, nt = w >> 1;
  f32x4 acc = {0, 0, 0, 0};
  for (int c = 0; c < 2; ++c) {
    int obase = c * 288;
    __syncthreads();
    for (int i = threadIdx.x; i < 32 * 288; i += 256) {
      // thread-consecutive co -> 128 B contiguous dz2t reads per o
      int o = obase + i / 32, co = i % 32;
      dzl[co * MC2FW_LD + (o - obase)] = dzb[o * 64 + ch * 32 + co];
    }
    for (int i = threadIdx.x; i < 32 * 288; i += 256) {
      int nr = i / 288, o = obase + i % 288;
      int n = nb * 32 + nr;
      int ci = n / 9, rem = n % 9, kh = rem / 3, kw = rem % 3;
      int yy = o / 24, xx = o % 24;
      imt[nr * MC2FW_LD + (o - obase)] =
          a1b[ci * 676 + (yy + kh) * 26 + xx + kw];
    }
    __syncthreads();
    #pragma unroll 8
    for (int k0 = 0; k0 < 288; k0 += 4) {
      int o = k0 + kc;
      float a = dzl[(csub * 16 + il) * MC2FW_LD + o];
      float bv = imt[(nt * 16 + il) * MC2FW_LD + o];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
  }
  float* out = slab + g * 18432;
  int orow = ch * 32 + csub * 16 + (lane >> 4) * 4;
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    out[(orow + r) * 288 + nb * 32 + nt * 16 + il] = acc[r];
}

__global__ void k_conv2_bwd_w_fold_mb(const float* __restrict__ slab,
                                      float* __restrict__ grads, long long P,
                                      long long ow2, int bs, int K) {
  int per_k = (18432 + FBLK - 1) / FBLK;
  int k = blockIdx.x / per_k;
  int i = (blockIdx.x % per_k) * FBLK + threadIdx.x;
  if (i >= 18432) return;
  const float* slk = slab + (long long)k * bs * 18432;
  float s = 0.f;
  for (int b = 0; b < bs; ++b) s += slk[(long long)b * 18432 + i];
  grads[(long long)k * P + ow2 + i] = s;
}

__global__ void k_conv2_bwd_b_mb(const float* __restrict__ dz2,
                                 float* __restrict__ grads, long long P,
                                 long long ob2, int bs, int K) {
  int k = blockIdx.x / 64;
  int co = blockIdx.x % 64;
  const float* dzk = dz2 + (long long)k * bs * 36864;
  float s = 0.f;
  for (int t = threadIdx.x; t < bs * 576; t += blockDim.x) {
    int o = t % 576, b = t / 576;
    s += dzk[((long long)b * 64 + co) * 576 + o];
  }
  for (int d = 32; d > 0; d >>= 1) s += __shfl_down(s, d, 64);
  __shared__ float lds[4];
  if ((threadIdx.x & 63) == 0) lds[threadIdx.x >> 6] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tt = 0.f;
    for (int q = 0; q < (int)blockDim.x / 64; ++q) tt += lds[q];
    grads[(long long)k * P + ob2 + co] = tt;
  }
}

// conv2 backward data (f32 MFMA): 32-row materialized-im2col tiles
// (74.9 KB LDS -> 2 blocks/CU), one accumulator per wave (wave w:
// m-subtile w&1, ci-tile w>>1), B streamed from the per-batch global
// w2rot layout with lane-contiguous 64 B groups.  grid = G * 22.
__global__ __launch_bounds__(256)
void k_conv2_bwd_x_mfma_mb(const float* __restrict__ dz2,
                           const float* __restrict__ w2rot_stack,
                           const float* __restrict__ a1, int bs, int K,
                           float* __restrict__ dz1) {
  __shared__ float imc[32 * MC2F_LD];
  long long g = blockIdx.x / 22;
  int mt = blockIdx.x % 22;
  int k = (int)(g / bs);
  const float* dzb = dz2 + g * 36864;
  const float* w2rot = w2rot_stack + (long long)k * 18432;
  for (int i = threadIdx.x; i < 32 * 576; i += 256) {
    int mr = i / 576, kk = i % 576;
    int m = mt * 32 + mr, p = m / 26, q = m % 26;
    int co = kk / 9, rem = kk % 9, kh = rem / 3, kw = rem % 3;
    int y = p - kh, x = q - kw;
    imc[mr * MC2F_LD + kk] =
        (m < 676 && y >= 0 && y < 24 && x >= 0 && x < 24)
            ? dzb[co * 576 + y * 24 + x] : 0.f;
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc = lane >> 4;
  int msub = w & 1, nt = w >> 1;
  f32x4 acc = {0, 0, 0, 0};
  #pragma unroll 8
  for (int k0 = 0; k0 < 576; k0 += 4) {
    int kk = k0 + kc;
    float a = imc[(msub * 16 + il) * MC2F_LD + kk];
    float bv = w2rot[(long long)kk * 32 + nt * 16 + il];
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
  }
  int om = mt * 32 + msub * 16 + (lane >> 4) * 4;
  int ci = nt * 16 + il;
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    if (om + r >= 676) continue;
    long long o = (g * 32 + ci) * 676 + om + r;
    dz1[o] = a1[o] > 0.f ? acc[r] : 0.f;
  }
}

__global__ void k_conv1_bwd_w_mb(const float* __restrict__ x,
                                 const float* __restrict__ dz1,
                                 float* __restrict__ grads, long long P,
                                 long long ow1, long long ob1, int bs,
                                 int K) {
  int k = blockIdx.x / 32;
  int co = blockIdx.x % 32;
  const float* xk = x + (long long)k * bs * 784;
  const float* dzk = dz1 + (long long)k * bs * 21632;
  float acc[9] = {0, 0, 0, 0, 0, 0, 0, 0, 0};
  float accb = 0.f;
  for (int t = threadIdx.x; t < bs * 676; t += blockDim.x) {
    int o = t % 676, b = t / 676;
    int xx = o % 26, yy = o / 26;
    float d = dzk[((long long)b * 32 + co) * 676 + o];
    const float* xp = xk + b * 784 + yy * 28 + xx;
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh)
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw)
        acc[kh * 3 + kw] = fmaf(xp[kh * 28 + kw], d, acc[kh * 3 + kw]);
    accb += d;
  }
  __shared__ float lds[16 * 10];
  int n_waves = blockDim.x / 64;
  int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  #pragma unroll
  for (int kk = 0; kk < 9; ++kk) {
    float s = acc[kk];
    for (int d = 32; d > 0; d >>= 1) s += __shfl_down(s, d, 64);
    if (lane == 0) lds[wave * 10 + kk] = s;
  }
  float sb = accb;
  for (int d = 32; d > 0; d >>= 1) sb += __shfl_down(sb, d, 64);
  if (lane == 0) lds[wave * 10 + 9] = sb;
  __syncthreads();
  if (threadIdx.x < 9) {
    float s = 0.f;
    for (int w = 0; w < n_waves; ++w) s += lds[w * 10 + threadIdx.x];
    grads[(long long)k * P + ow1 + co * 9 + threadIdx.x] = s;
  }
  if (threadIdx.x == 9) {
    float s = 0.f;
    for (int w = 0; w < n_waves; ++w) s += lds[w * 10 + 9];
    grads[(long long)k * P + ob1 + co] = s;
  }
}

// per-client clip + sufficient stats + SGD over the gradient stacks.
// Stage 1: per-(k, chunk) partial sum/sumsq -> f64 atomics into acc[2k]
// (few hundred atomics per client per step).  Stage 2: per-client scale
// + stats accumulate + SGD in one elementwise pass.
// one block per (client, contiguous chunk): block-local tree reduce,
// ONE f64 atomicAdd pair per block (~K * ceil(P/65536) atomics per step)
#define SUMSQ_CHUNK 32768
__global__ void k_sumsq_mb(const float* __restrict__ grads, long long P,
                           int blocks_per_k, double* __restrict__ acc) {
  int k = blockIdx.x / blocks_per_k;
  int c = blockIdx.x % blocks_per_k;
  long long lo = (long long)c * SUMSQ_CHUNK;
  long long hi = lo + SUMSQ_CHUNK;
  if (hi > P) hi = P;
  const float* g = grads + (long long)k * P;
  // float4 loads + independent accumulator pairs break the dependent
  // f64-add chain (was ~74 us per step at 190 underfilled blocks)
  double s0 = 0.0, s1 = 0.0, s2 = 0.0, s3 = 0.0;
  double q0 = 0.0, q1 = 0.0, q2 = 0.0, q3 = 0.0;
  // scalar loads (the k*P segment bases are not 16 B aligned); the win
  // is the 4 independent f64 accumulator chains
  long long i = lo + (long long)threadIdx.x * 4;
  for (; i + 3 < hi; i += (long long)blockDim.x * 4) {
    double a = g[i], b = g[i + 1], cc = g[i + 2], d = g[i + 3];
    s0 += a; q0 += a * a;
    s1 += b; q1 += b * b;
    s2 += cc; q2 += cc * cc;
    s3 += d; q3 += d * d;
  }
  for (long long j = i; j < hi; ++j) {  // ragged tail (P % 4)
    double v = (double)g[j];
    s0 += v; q0 += v * v;
  }
  double fs = (s0 + s1) + (s2 + s3);
  double fq = (q0 + q1) + (q2 + q3);
  {
  }
  for (int d = 32; d > 0; d >>= 1) {
    fs += __shfl_down(fs, d, 64);
    fq += __shfl_down(fq, d, 64);
  }
  __shared__ double lds[2 * 4];
  int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  if (lane == 0) { lds[2 * wave] = fs; lds[2 * wave + 1] = fq; }
  __syncthreads();
  if (threadIdx.x == 0) {
    double ts = 0.0, tq = 0.0;
    for (int w = 0; w < (int)blockDim.x / 64; ++w) {
      ts += lds[2 * w]; tq += lds[2 * w + 1];
    }
    atomicAdd(acc + 2 * k, ts);
    atomicAdd(acc + 2 * k + 1, tq);
  }
}

__global__ void k_clip_sgd_mb(float* __restrict__ params,
                              float* __restrict__ grads, long long P, int K,
                              const double* __restrict__ acc, float max_norm,
                              float eps, const float* __restrict__ lr_t,
                              float* __restrict__ stats_out) {
  long long total = (long long)K * P;
  float lr = lr_t[0];
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i / P);
    float norm = (float)sqrt(acc[2 * k + 1]);
    float scale = (max_norm > 0.f && norm > max_norm)
                      ? max_norm / (norm + eps) : 1.f;
    float gv = grads[i] * scale;
    grads[i] = gv;
    params[i] -= lr * gv;
  }
}

// per-client clipped-stats accumulate: stats_out[2k] += scale*sum,
// stats_out[2k+1] += scale^2*sumsq (one thread per client)
__global__ void k_stats_mb(const double* __restrict__ acc, int K,
                           float max_norm, float eps,
                           float* __restrict__ stats_out) {
  int k = blockIdx.x * blockDim.x + threadIdx.x;
  if (k >= K) return;
  float norm = (float)sqrt(acc[2 * k + 1]);
  float scale = (max_norm > 0.f && norm > max_norm)
                    ? max_norm / (norm + eps) : 1.f;
  stats_out[2 * k] += scale * (float)acc[2 * k];
  stats_out[2 * k + 1] += scale * scale * (float)acc[2 * k + 1];
}

// weighted pseudo-gradients + deterministic accumulate into round_accum
__global__ void k_pseudo_grad_mb(float* __restrict__ grads,
                                 const float* __restrict__ server,
                                 const float* __restrict__ params,
                                 const float* __restrict__ weights,
                                 long long P, int K) {
  long long total = (long long)K * P;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i / P);
    grads[i] = (server[i % P] - params[i]) * weights[k];
  }
}

__global__ void k_accum_mb(float* __restrict__ round_accum,
                           const float* __restrict__ grads, long long P,
                           int K) {
  for (long long p = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       p < P; p += (long long)gridDim.x * blockDim.x) {
    float s = round_accum[p];
    for (int k = 0; k < K; ++k) s += grads[(long long)k * P + p];
    round_accum[p] = s;
  }
}

// ---------------------------------------------------------------------------
// bf16 mega variants (mixed precision, fp32 master — same recipe as the
// per-client bf16 kernels in fused_cnn.hip: operands materialized as
// k-contiguous bf16 LDS tiles with padded strides, f32 accumulators,
// MFMA v_mfma_f32_16x16x32_bf16; the inner loops read ONLY LDS, so the
// L2-latency bound of the f32 variants does not apply).
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8m;

__device__ __forceinline__ bf16x8m ld_bf16x8m(const __bf16* p) {
  return *reinterpret_cast<const bf16x8m*>(p);
}

// [k][ci][k'=(co,kh,kw)] bf16 layout for the bwd-data B operand (built
// once per batch into the w2rot_stack buffer reinterpreted as bf16)
__global__ void k_w2rotbf_mb(const float* __restrict__ params, long long P,
                             long long ow2, int K,
                             __bf16* __restrict__ w2rotbf) {
  long long total = (long long)K * 18432;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i / 18432), r = (int)(i % 18432);
    int co = r / 288, rem9 = r % 288;
    int ci = rem9 / 9, rem = rem9 % 9;
    w2rotbf[(long long)k * 18432 + (long long)ci * 576 + co * 9 + rem] =
        (__bf16)params[(long long)k * P + ow2 + r];
  }
}

#define MC2_LD 296
__global__ __launch_bounds__(256)
void k_conv2_fwd_mfma_mb_bf16(const float* __restrict__ a1,
                              const float* __restrict__ params, long long P,
                              long long ow2, long long ob2, int bs, int K,
                              float* __restrict__ r2) {
  __shared__ __bf16 imc[64 * MC2_LD];
  __shared__ __bf16 wb[64 * MC2_LD];
  long long g = blockIdx.x / 9;
  int mt = blockIdx.x % 9;
  int k = (int)(g / bs);
  const float* a1b = a1 + g * 21632;
  const float* w2 = params + (long long)k * P + ow2;
  const float* b2 = params + (long long)k * P + ob2;
  for (int i = threadIdx.x; i < 64 * 288; i += 256) {
    int mr = i / 288, kk = i % 288;
    int m = mt * 64 + mr, yy = m / 24, xx = m % 24;
    int ci = kk / 9, rem = kk % 9, kh = rem / 3, kw = rem % 3;
    imc[mr * MC2_LD + kk] = (__bf16)a1b[ci * 676 + (yy + kh) * 26 + xx + kw];
    wb[mr * MC2_LD + kk] = (__bf16)w2[i];  // mr doubles as co
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  f32x4 acc[4] = {f32x4{0,0,0,0}, f32x4{0,0,0,0},
                  f32x4{0,0,0,0}, f32x4{0,0,0,0}};
  for (int k0 = 0; k0 < 288; k0 += 32) {
    bf16x8m a = ld_bf16x8m(&imc[(w * 16 + il) * MC2_LD + k0 + kc8]);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      bf16x8m bv = ld_bf16x8m(&wb[(nt * 16 + il) * MC2_LD + k0 + kc8]);
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
      r2[(g * 64 + co) * 576 + om + r] = v > 0.f ? v : 0.f;
    }
}

#define MC2B_LD 584
// 32-row m-tiles (imc 37.4 KB -> 4 blocks/CU of 4 waves hide the staging
// latency that bound the 64-row form to ~340 us); the B operand streams
// from the pre-built global w2rotbf (18 16-B reads per lane, unrolled).
// Wave w: m-subtile (w & 1), ci-tile (w >> 1).
__global__ __launch_bounds__(256)
void k_conv2_bwd_x_mfma_mb_bf16(const float* __restrict__ dz2,
                                const __bf16* __restrict__ w2rotbf,
                                int bs, int K,
                                const float* __restrict__ a1,
                                float* __restrict__ dz1) {
  __shared__ __bf16 imc[32 * MC2B_LD];
  long long g = blockIdx.x / 22;
  int mt = blockIdx.x % 22;
  int k = (int)(g / bs);
  const float* dzb = dz2 + g * 36864;
  const __bf16* wtk = w2rotbf + (long long)k * 18432;
  for (int i = threadIdx.x; i < 32 * 576; i += 256) {
    int mr = i / 576, kk = i % 576;
    int m = mt * 32 + mr, p = m / 26, q = m % 26;
    int co = kk / 9, rem = kk % 9, kh = rem / 3, kw = rem % 3;
    int y = p - kh, x = q - kw;
    imc[mr * MC2B_LD + kk] =
        (__bf16)((m < 676 && y >= 0 && y < 24 && x >= 0 && x < 24)
                     ? dzb[co * 576 + y * 24 + x] : 0.f);
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  int msub = w & 1, nt = w >> 1;
  f32x4 acc = {0, 0, 0, 0};
  #pragma unroll 3
  for (int k0 = 0; k0 < 576; k0 += 32) {
    bf16x8m a = ld_bf16x8m(&imc[(msub * 16 + il) * MC2B_LD + k0 + kc8]);
    bf16x8m bv = ld_bf16x8m(&wtk[(long long)(nt * 16 + il) * 576
                                 + k0 + kc8]);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc, 0, 0, 0);
  }
  int om = mt * 32 + msub * 16 + (lane >> 4) * 4;
  int ci = nt * 16 + il;
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    if (om + r >= 676) continue;
    long long o = (g * 32 + ci) * 676 + om + r;
    dz1[o] = a1[o] > 0.f ? acc[r] : 0.f;
  }
}

// block = 32 co-rows x 32 n-cols over K=576 (dzb16 37.4 KB + imt
// 37.4 KB -> 2 blocks/CU); grid = G * 2(co) * 9(n).  Wave w: co-subtile
// (w & 1), n-subtile (w >> 1).
__global__ __launch_bounds__(256)
void k_conv2_bwd_w_mfma_mb_bf16(const float* __restrict__ dz2,
                                const float* __restrict__ a1, int bs, int K,
                                float* __restrict__ slab) {
  __shared__ __bf16 dzb16[32 * MC2B_LD];
  __shared__ __bf16 imt[32 * MC2B_LD];
  int blk = blockIdx.x % 18;
  int ch = blk / 9;         // co half (0: rows 0-31, 1: rows 32-63)
  int nb = blk % 9;         // 32-col n-block
  long long g = blockIdx.x / 18;
  const float* dzb = dz2 + g * 36864 + (long long)ch * 32 * 576;
  const float* a1b = a1 + g * 21632;
  for (int i = threadIdx.x; i < 32 * 576; i += 256) {
    int co = i / 576;
    dzb16[co * MC2B_LD + (i % 576)] = (__bf16)dzb[i];
  }
  for (int i = threadIdx.x; i < 32 * 576; i += 256) {
    int nr = i / 576, o = i % 576;
    int n = nb * 32 + nr;
    int ci = n / 9, rem = n % 9, kh = rem / 3, kw = rem % 3;
    int yy = o / 24, xx = o % 24;
    imt[nr * MC2B_LD + o] = (__bf16)a1b[ci * 676 + (yy + kh) * 26 + xx + kw];
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  int csub = w & 1, nt = w >> 1;
  f32x4 acc = {0, 0, 0, 0};
  #pragma unroll 3
  for (int k0 = 0; k0 < 576; k0 += 32) {
    bf16x8m a = ld_bf16x8m(&dzb16[(csub * 16 + il) * MC2B_LD + k0 + kc8]);
    bf16x8m bv = ld_bf16x8m(&imt[(nt * 16 + il) * MC2B_LD + k0 + kc8]);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc, 0, 0, 0);
  }
  float* out = slab + g * 18432;
  int orow = ch * 32 + csub * 16 + (lane >> 4) * 4;
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    out[(orow + r) * 288 + nb * 32 + nt * 16 + il] = acc[r];
}

#define MFC1B_SPLIT 36
#define MFC1B_CH 256
#define MFC1B_LD 264
__global__ __launch_bounds__(256)
void k_fc1_fwd_mfma_mb_bf16(const float* __restrict__ a2,
                            const float* __restrict__ params, long long P,
                            long long ow3, int bs, int K,
                            float* __restrict__ slab) {
  __shared__ __bf16 lw[128 * MFC1B_LD];
  __shared__ __bf16 la[32 * MFC1B_LD];
  int k = blockIdx.x / MFC1B_SPLIT;
  int s = blockIdx.x % MFC1B_SPLIT;
  int k_base = s * MFC1B_CH;
  const float* w3 = params + (long long)k * P + ow3;
  const float* a2k = a2 + (long long)k * bs * 9216;
  for (int i = threadIdx.x; i < 128 * MFC1B_CH; i += 256) {
    int row = i / MFC1B_CH, kk = i % MFC1B_CH;
    lw[row * MFC1B_LD + kk] = (__bf16)w3[(long long)row * 9216 + k_base + kk];
  }
  for (int i = threadIdx.x; i < 32 * MFC1B_CH; i += 256) {
    int bu = i / MFC1B_CH, kk = i % MFC1B_CH;
    la[bu * MFC1B_LD + kk] =
        (__bf16)(bu < bs ? a2k[(long long)bu * 9216 + k_base + kk] : 0.f);
  }
  __syncthreads();
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int il = lane & 15, kc8 = (lane >> 4) * 8;
  f32x4 acc[2][2] = {{f32x4{0,0,0,0}, f32x4{0,0,0,0}},
                     {f32x4{0,0,0,0}, f32x4{0,0,0,0}}};
  for (int k0 = 0; k0 < MFC1B_CH; k0 += 32) {
    bf16x8m a0 = ld_bf16x8m(&lw[(w * 32 + il) * MFC1B_LD + k0 + kc8]);
    bf16x8m a1v = ld_bf16x8m(&lw[(w * 32 + 16 + il) * MFC1B_LD + k0 + kc8]);
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      bf16x8m bv = ld_bf16x8m(&la[(u * 16 + il) * MFC1B_LD + k0 + kc8]);
      acc[0][u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bv, acc[0][u], 0, 0, 0);
      acc[1][u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1v, bv, acc[1][u], 0, 0, 0);
    }
  }
  float* slk = slab + (long long)k * MFC1B_SPLIT * bs * 128;
  #pragma unroll
  for (int tt = 0; tt < 2; ++tt)
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      int bu = u * 16 + il;
      if (bu >= bs) continue;
      int j = w * 32 + tt * 16 + (lane >> 4) * 4;
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        slk[((long long)s * bs + bu) * 128 + j + r] = acc[tt][u][r];
    }
}

__global__ void k_fc1_fwd_reduce_mb_bf16(const float* __restrict__ slab,
                                         const float* __restrict__ params,
                                         long long P, long long ob3, int bs,
                                         int K, float p2,
                                         const long long* __restrict__ seeds,
                                         unsigned long long offset,
                                         float* __restrict__ z3,
                                         float* __restrict__ a3,
                                         unsigned char* __restrict__ m3) {
  long long total = (long long)K * bs * 128;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    long long g = i / 128;
    int j = (int)(i % 128);
    int k = (int)(g / bs), b = (int)(g % bs);
    const float* slk = slab + (long long)k * MFC1B_SPLIT * bs * 128;
    float t = params[(long long)k * P + ob3 + j];
    for (int s = 0; s < MFC1B_SPLIT; ++s)
      t += slk[((long long)s * bs + b) * 128 + j];
    z3[i] = t;
    float r = t > 0.f ? t : 0.f;
    unsigned char keep = 1;
    if (p2 > 0.f) {
      long long li = (long long)b * 128 + j;
      hiprandStatePhilox4_32_10_t st;
      hiprand_init((unsigned long long)seeds[k] ^ 0x9e3779b97f4a7c15ull,
                   (unsigned long long)li, offset, &st);
      keep = hiprand_uniform(&st) >= p2;
    }
    m3[i] = keep;
    a3[i] = keep ? r / (1.f - p2) : 0.f;
  }
}


// ---------------------------------------------------------------------------
// host driver: one launch set per batch-step for ALL K clients
// ---------------------------------------------------------------------------
extern "C" void launch_cnn_round_mega(
    const float* shard_x, const long long* shard_y,
    const long long* orders_dev,
    const long long* row_bases_dev, const long long* order_offs_dev,
    const long long* counts_dev, const long long* counts_host,
    const float* weights_dev, const long long* seeds_dev,
    int K, int bs, int C,
    const float* server_params, float* params_stack, float* grads_stack,
    float* round_accum,
    float* xb, float* a1, float* r2, float* a2, float* z3, float* a3,
    float* dlogits, float* dz3, float* da2, float* dz2, float* dz1,
    float* w2t_stack, float* w2rot_stack, float* slab,
    int* yb, unsigned char* pidx, unsigned char* m2, unsigned char* m3,
    double* acc2k,
    const float* lr_t, float max_norm, float p1, float p2,
    float* stats_out, float* loss_out, hipStream_t s, int use_bf16) {
  MegaOffsets o = mega_offsets(C);
  long long P = o.total;
  int G = K * bs;
  int max_batches = 0;
  for (int k = 0; k < K; ++k) {
    int nb = (int)((counts_host[k] + bs - 1) / bs);
    if (nb > max_batches) max_batches = nb;
  }
  long long kp = (long long)K * P;
  int gkp = (int)((kp + FBLK - 1) / FBLK); if (gkp > 4096) gkp = 4096;
  int gp = (int)((P + FBLK - 1) / FBLK); if (gp > 2048) gp = 2048;
  hipLaunchKernelGGL(k_copy_stack, dim3(gkp), dim3(FBLK), 0, s,
                     params_stack, server_params, P, K);
  for (int t = 0; t < max_batches; ++t) {
    unsigned long long off = (unsigned long long)t;
    hipLaunchKernelGGL(k_gather_mb, dim3((G * 784 + FBLK - 1) / FBLK),
                       dim3(FBLK), 0, s, shard_x, shard_y, orders_dev,
                       row_bases_dev, order_offs_dev, counts_dev, t, bs, K,
                       xb, yb);
    hipLaunchKernelGGL(k_conv1_fwd_mb,
                       dim3((int)(((long long)G * 21632 + FBLK - 1) / FBLK)),
                       dim3(FBLK), 0, s, xb, params_stack, P, o.w1, o.b1,
                       bs, K, a1);
    if (use_bf16) {
      hipLaunchKernelGGL(k_w2rotbf_mb,
                         dim3((int)(((long long)K * 18432 + FBLK - 1) / FBLK)),
                         dim3(FBLK), 0, s, params_stack, P, o.w2, K,
                         reinterpret_cast<__bf16*>(w2rot_stack));
      hipLaunchKernelGGL(k_conv2_fwd_mfma_mb_bf16, dim3(G * 9), dim3(FBLK),
                         0, s, a1, params_stack, P, o.w2, o.b2, bs, K, r2);
    } else {
      hipLaunchKernelGGL(k_w2_layouts_mb,
                         dim3((int)(((long long)K * 18432 + FBLK - 1) / FBLK)),
                         dim3(FBLK), 0, s, params_stack, P, o.w2, K,
                         w2t_stack, w2rot_stack);
      hipLaunchKernelGGL(k_conv2_fwd_mfma_mb, dim3(G * 9), dim3(FBLK), 0, s,
                         a1, w2t_stack, params_stack, P, o.b2, bs, K, r2);
    }
    hipLaunchKernelGGL(k_pool_drop_fwd_mb,
                       dim3((int)(((long long)G * 9216 + FBLK - 1) / FBLK)),
                       dim3(FBLK), 0, s, r2, bs, K, p1, seeds_dev, off,
                       a2, pidx, m2);
    if (use_bf16) {
      hipLaunchKernelGGL(k_fc1_fwd_mfma_mb_bf16, dim3(K * 36), dim3(FBLK),
                         0, s, a2, params_stack, P, o.w3, bs, K, slab);
      hipLaunchKernelGGL(k_fc1_fwd_reduce_mb_bf16,
                         dim3((int)(((long long)G * 128 + FBLK - 1) / FBLK)),
                         dim3(FBLK), 0, s, slab, params_stack, P, o.b3, bs,
                         K, p2, seeds_dev, off, z3, a3, m3);
    } else {
      hipLaunchKernelGGL(k_fc1_fwd_mfma_mb, dim3(K * FC1M_SPLIT), dim3(FBLK),
                         0, s, a2, params_stack, P, o.w3, bs, K, slab);
      hipLaunchKernelGGL(k_fc1_fwd_reduce_mb,
                         dim3((int)(((long long)G * 128 + FBLK - 1) / FBLK)),
                         dim3(FBLK), 0, s, slab, params_stack, P, o.b3, bs,
                         K, p2, seeds_dev, off, z3, a3, m3);
    }
    hipLaunchKernelGGL(k_fc2_loss_fwd_mb, dim3(G), dim3(FBLK),
                       C * (int)sizeof(float), s, a3, params_stack, P, o.w4,
                       o.b4, counts_dev, t, bs, K, C, yb, dlogits, loss_out);
    int perk_fc2 = (C * 128 + FBLK - 1) / FBLK;
    hipLaunchKernelGGL(k_fc2_bwd_w_mb, dim3(K * perk_fc2), dim3(FBLK), 0, s,
                       dlogits, a3, grads_stack, P, o.w4, o.b4, bs, K, C);
    hipLaunchKernelGGL(k_fc2_bwd_x_mb,
                       dim3((int)(((long long)G * 128 + FBLK - 1) / FBLK)),
                       dim3(FBLK), 0, s, dlogits, params_stack, P, o.w4,
                       z3, m3, bs, K, C, p2, dz3);
    hipLaunchKernelGGL(k_fc1_bwd_w_mfma_mb, dim3(K * 144), dim3(FBLK), 0, s,
                       dz3, a2, grads_stack, P, o.w3, bs, K);
    hipLaunchKernelGGL(k_fc1_bwd_b_mb, dim3(K), dim3(128), 0, s,
                       dz3, grads_stack, P, o.b3, bs, K);
    hipLaunchKernelGGL(k_fc1_bwd_x_mfma_mb, dim3(K * 144), dim3(FBLK), 0, s,
                       dz3, params_stack, P, o.w3, bs, K, da2);
    hipLaunchKernelGGL(k_pool_drop_bwd_mb,
                       dim3((int)(((long long)G * 9216 + FBLK - 1) / FBLK)),
                       dim3(FBLK), 0, s, da2, pidx, m2, r2, bs, K, p1, dz2);
    if (use_bf16) {
      hipLaunchKernelGGL(k_conv2_bwd_w_mfma_mb_bf16, dim3(18 * G), dim3(FBLK),
                         0, s, dz2, a1, bs, K, slab);
    } else {
      hipLaunchKernelGGL(k_dz2_transpose_mb,
                         dim3((int)(((long long)G * 36864 + FBLK - 1) / FBLK)),
                         dim3(FBLK), 0, s, dz2, bs, K, r2);  // r2 free now
      hipLaunchKernelGGL(k_conv2_bwd_w_mfma_mb, dim3(18 * G), dim3(FBLK),
                         0, s, r2, a1, bs, K, slab);
    }
    int perk_fold = (18432 + FBLK - 1) / FBLK;
    hipLaunchKernelGGL(k_conv2_bwd_w_fold_mb, dim3(K * perk_fold),
                       dim3(FBLK), 0, s, slab, grads_stack, P, o.w2, bs, K);
    hipLaunchKernelGGL(k_conv2_bwd_b_mb, dim3(K * 64), dim3(FBLK), 0, s,
                       dz2, grads_stack, P, o.b2, bs, K);
    if (use_bf16)
      hipLaunchKernelGGL(k_conv2_bwd_x_mfma_mb_bf16, dim3(G * 22), dim3(FBLK),
                         0, s, dz2,
                         reinterpret_cast<const __bf16*>(w2rot_stack),
                         bs, K, a1, dz1);
    else
      hipLaunchKernelGGL(k_conv2_bwd_x_mfma_mb, dim3(G * 22), dim3(FBLK),
                         0, s, dz2, w2rot_stack, a1, bs, K, dz1);
    hipLaunchKernelGGL(k_conv1_bwd_w_mb, dim3(K * 32), dim3(1024), 0, s,
                       xb, dz1, grads_stack, P, o.w1, o.b1, bs, K);
    hipMemsetAsync(acc2k, 0, 2 * K * sizeof(double), s);
    int blocks_per_k = (int)((P + SUMSQ_CHUNK - 1) / SUMSQ_CHUNK);
    hipLaunchKernelGGL(k_sumsq_mb, dim3(K * blocks_per_k), dim3(FBLK), 0, s,
                       grads_stack, P, blocks_per_k, acc2k);
    hipLaunchKernelGGL(k_stats_mb, dim3((K + 63) / 64), dim3(64), 0, s,
                       acc2k, K, max_norm, 1e-6f, stats_out);
    hipLaunchKernelGGL(k_clip_sgd_mb, dim3(gkp), dim3(FBLK), 0, s,
                       params_stack, grads_stack, P, K, acc2k, max_norm,
                       1e-6f, lr_t, stats_out);
  }
  hipLaunchKernelGGL(k_pseudo_grad_mb, dim3(gkp), dim3(FBLK), 0, s,
                     grads_stack, server_params, params_stack, weights_dev,
                     P, K);
  hipLaunchKernelGGL(k_accum_mb, dim3(gp), dim3(FBLK), 0, s,
                     round_accum, grads_stack, P, K);
}


// ---------------------------------------------------------------------------
// standalone entries for K-stacked flat arenas (reused by the
// Shakespeare mega round's graph-captured epoch): per-client clip +
// stats + SGD, and weighted pseudo-grad + deterministic accumulate.
// ---------------------------------------------------------------------------
extern "C" void launch_mega_clip_sgd(float* params_stack, float* grads_stack,
                                     long long P, int K, double* acc2k,
                                     float max_norm, const float* lr_t,
                                     float* stats_out, hipStream_t s) {
  hipMemsetAsync(acc2k, 0, 2 * K * sizeof(double), s);
  int blocks_per_k = (int)((P + SUMSQ_CHUNK - 1) / SUMSQ_CHUNK);
  hipLaunchKernelGGL(k_sumsq_mb, dim3(K * blocks_per_k), dim3(FBLK), 0, s,
                     grads_stack, P, blocks_per_k, acc2k);
  hipLaunchKernelGGL(k_stats_mb, dim3((K + 63) / 64), dim3(64), 0, s,
                     acc2k, K, max_norm, 1e-6f, stats_out);
  long long kp = (long long)K * P;
  int gkp = (int)((kp + FBLK - 1) / FBLK); if (gkp > 4096) gkp = 4096;
  hipLaunchKernelGGL(k_clip_sgd_mb, dim3(gkp), dim3(FBLK), 0, s,
                     params_stack, grads_stack, P, K, acc2k, max_norm,
                     1e-6f, lr_t, stats_out);
}

extern "C" void launch_mega_pseudo_accum(float* grads_stack,
                                         const float* server,
                                         const float* params_stack,
                                         const float* weights_dev,
                                         float* round_accum, long long P,
                                         int K, hipStream_t s) {
  long long kp = (long long)K * P;
  int gkp = (int)((kp + FBLK - 1) / FBLK); if (gkp > 4096) gkp = 4096;
  int gp = (int)((P + FBLK - 1) / FBLK); if (gp > 2048) gp = 2048;
  hipLaunchKernelGGL(k_pseudo_grad_mb, dim3(gkp), dim3(FBLK), 0, s,
                     grads_stack, server, params_stack, weights_dev, P, K);
  hipLaunchKernelGGL(k_accum_mb, dim3(gp), dim3(FBLK), 0, s,
                     round_accum, grads_stack, P, K);
}
