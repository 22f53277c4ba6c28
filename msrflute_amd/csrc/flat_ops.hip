// CDNA4 (gfx950) flat-arena kernels for msrflute_amd.
//
// Every op here runs over one contiguous fp32 buffer (the parameter /
// gradient arena, msrflute_amd/ops/arena.py).  These replace the reference
// implementation's per-tensor Python loops:
// pseudo-gradient, weighted accumulate, fused gradient statistics, norm
// clipping, Philox Gaussian DP noise, quantization binning and fused
// optimizer steps.
//
// Design notes (per the CDNA4 programming guide):
//  * memory-bound elementwise ops: float4 (16 B/lane) vectorized loads,
//    256-thread blocks, grid-stride with the grid capped at 2048 blocks
//    (>> 256 workgroups fills all 8 XCDs; cap leaves scheduler room);
//  * reductions: 64-wide wavefront __shfl_down tree -> LDS across the
//    block's 4 waves -> one global double atomicAdd per block (CDNA has
//    native f64 global atomics);
//  * noise: Philox4x32-10 keyed by (seed, element-block index) so the
//    stream is independent of grid configuration — bitwise reproducible
//    for a fixed (seed, offset);
//  * all kernels are launched on the current torch stream by bindings.cpp.

#include <hip/hip_runtime.h>
#include <hiprand/hiprand_kernel.h>

#define WAVE 64
#define BLOCK 256
#define MAX_BLOCKS 2048
// Reductions funnel into 2 f64 atomics per block on ONE address; the
// serialized atomic chain dominates past a few hundred blocks (measured
// 30 us for a 1.2M-param arena at 1182 blocks), so cap reduction grids
// low and loop more per thread instead.
#define MAX_BLOCKS_REDUCE 192

static inline int grid_for(long long n_vec) {
  long long b = (n_vec + BLOCK - 1) / BLOCK;
  if (b > MAX_BLOCKS) b = MAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}

static inline int grid_for_reduce(long long n_vec) {
  long long b = (n_vec + BLOCK - 1) / BLOCK;
  if (b > MAX_BLOCKS_REDUCE) b = MAX_BLOCKS_REDUCE;
  if (b < 1) b = 1;
  return (int)b;
}

// ---------------------------------------------------------------------------
// Elementwise ops (vectorized float4 main body + scalar tail)
// ---------------------------------------------------------------------------

__global__ void k_pseudo_grad(float* __restrict__ out,
                              const float* __restrict__ ws,
                              const float* __restrict__ wt,
                              float weight, long long n) {
  long long n4 = n >> 2;
  const float4* ws4 = reinterpret_cast<const float4*>(ws);
  const float4* wt4 = reinterpret_cast<const float4*>(wt);
  float4* out4 = reinterpret_cast<float4*>(out);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = ws4[i], b = wt4[i], r;
    r.x = (a.x - b.x) * weight; r.y = (a.y - b.y) * weight;
    r.z = (a.z - b.z) * weight; r.w = (a.w - b.w) * weight;
    out4[i] = r;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    out[i] = (ws[i] - wt[i]) * weight;
}

__global__ void k_axpy(float* __restrict__ y, const float* __restrict__ x,
                       float alpha, long long n) {
  long long n4 = n >> 2;
  float4* y4 = reinterpret_cast<float4*>(y);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = y4[i], b = x4[i];
    a.x = fmaf(alpha, b.x, a.x); a.y = fmaf(alpha, b.y, a.y);
    a.z = fmaf(alpha, b.z, a.z); a.w = fmaf(alpha, b.w, a.w);
    y4[i] = a;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    y[i] = fmaf(alpha, x[i], y[i]);
}

__global__ void k_scale(float* __restrict__ x, float alpha, long long n) {
  long long n4 = n >> 2;
  float4* x4 = reinterpret_cast<float4*>(x);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = x4[i];
    a.x *= alpha; a.y *= alpha; a.z *= alpha; a.w *= alpha;
    x4[i] = a;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    x[i] *= alpha;
}

// ---------------------------------------------------------------------------
// Reductions: Σx and Σx² in one pass (double accumulation)
// ---------------------------------------------------------------------------

__device__ inline void block_reduce2_atomic(double s, double q,
                                            double* __restrict__ out2) {
  // wave-level tree over 64 lanes
  for (int d = WAVE / 2; d > 0; d >>= 1) {
    s += __shfl_down(s, d, WAVE);
    q += __shfl_down(q, d, WAVE);
  }
  __shared__ double lds_s[BLOCK / WAVE], lds_q[BLOCK / WAVE];
  int wave = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
  if (lane == 0) { lds_s[wave] = s; lds_q[wave] = q; }
  __syncthreads();
  if (threadIdx.x == 0) {
    double ts = 0, tq = 0;
    for (int w = 0; w < BLOCK / WAVE; ++w) { ts += lds_s[w]; tq += lds_q[w]; }
    atomicAdd(&out2[0], ts);
    atomicAdd(&out2[1], tq);
  }
}

__global__ void k_sum_sumsq(const float* __restrict__ x, long long n,
                            double* __restrict__ out2) {
  double s = 0.0, q = 0.0;
  long long n4 = n >> 2;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = x4[i];
    s += (double)a.x + (double)a.y + (double)a.z + (double)a.w;
    q += (double)a.x * a.x + (double)a.y * a.y +
         (double)a.z * a.z + (double)a.w * a.w;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    double v = x[i];
    s += v; q += v * v;
  }
  block_reduce2_atomic(s, q, out2);
}

// Two-stage variant: stage 1 writes one {Σ, Σ²} pair per block (no
// atomics — the serialized f64 atomic chain was the bottleneck at large
// grids); stage 2 folds the partials with one small block.
__global__ void k_sum_sumsq_partial(const float* __restrict__ x, long long n,
                                    double* __restrict__ partials) {
  double s = 0.0, q = 0.0;
  long long n4 = n >> 2;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = x4[i];
    s += (double)a.x + (double)a.y + (double)a.z + (double)a.w;
    q += (double)a.x * a.x + (double)a.y * a.y +
         (double)a.z * a.z + (double)a.w * a.w;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x) {
    double v = x[i];
    s += v; q += v * v;
  }
  for (int d = WAVE / 2; d > 0; d >>= 1) {
    s += __shfl_down(s, d, WAVE);
    q += __shfl_down(q, d, WAVE);
  }
  __shared__ double lds_s[BLOCK / WAVE], lds_q[BLOCK / WAVE];
  int wave = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
  if (lane == 0) { lds_s[wave] = s; lds_q[wave] = q; }
  __syncthreads();
  if (threadIdx.x == 0) {
    double ts = 0, tq = 0;
    for (int w = 0; w < BLOCK / WAVE; ++w) { ts += lds_s[w]; tq += lds_q[w]; }
    partials[2 * blockIdx.x] = ts;
    partials[2 * blockIdx.x + 1] = tq;
  }
}

__global__ void k_fold_partials(const double* __restrict__ partials,
                                int n_blocks, double* __restrict__ out2) {
  double s = 0.0, q = 0.0;
  for (int i = threadIdx.x; i < n_blocks; i += blockDim.x) {
    s += partials[2 * i];
    q += partials[2 * i + 1];
  }
  block_reduce2_atomic(s, q, out2);
}

// Scale x by min(1, max_norm/(sqrt(sumsq)+eps)); out2[1] holds Σx² from
// k_sum_sumsq; norm_out receives the pre-clip norm. One extra kernel, zero
// host synchronization (torch clip_grad_norm_ semantics).
__global__ void k_clip_apply(float* __restrict__ x, long long n,
                             const double* __restrict__ out2,
                             float max_norm, float eps,
                             float* __restrict__ norm_out) {
  double norm = sqrt(out2[1]);
  if (blockIdx.x == 0 && threadIdx.x == 0) *norm_out = (float)norm;
  float coef = (float)(max_norm / (norm + (double)eps));
  if (coef >= 1.0f) return;
  long long n4 = n >> 2;
  float4* x4 = reinterpret_cast<float4*>(x);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = x4[i];
    a.x *= coef; a.y *= coef; a.z *= coef; a.w *= coef;
    x4[i] = a;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    x[i] *= coef;
}

// Fused clip + sufficient stats: given out2 = {Σg, Σg²} from k_sum_sumsq,
// scale g by coef = min(1, max_norm/(‖g‖+eps)) and ACCUMULATE the
// post-clip stats {coef·Σg, coef²·Σg²} into stats_acc[0..1] — the whole
// per-batch clip+stats pipeline (K5+K6) costs ONE reduction pass.
__global__ void k_clip_apply_stats(float* __restrict__ x, long long n,
                                   const double* __restrict__ out2,
                                   float max_norm, float eps,
                                   float* __restrict__ stats_acc) {
  double norm = sqrt(out2[1]);
  float coef = (float)(max_norm / (norm + (double)eps));
  if (coef > 1.0f || max_norm <= 0.f) coef = 1.0f;
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    atomicAdd(&stats_acc[0], (float)(out2[0] * coef));
    atomicAdd(&stats_acc[1], (float)(out2[1] * coef * coef));
  }
  if (coef >= 1.0f) return;
  long long n4 = n >> 2;
  float4* x4 = reinterpret_cast<float4*>(x);
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    float4 a = x4[i];
    a.x *= coef; a.y *= coef; a.z *= coef; a.w *= coef;
    x4[i] = a;
  }
  long long tail = n4 << 2;
  for (long long i = tail + blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < n; i += (long long)gridDim.x * blockDim.x)
    x[i] *= coef;
}

// ---------------------------------------------------------------------------
// Philox Gaussian noise (DP): x[i] += sigma * N(0,1)
// ---------------------------------------------------------------------------

__global__ void k_add_gaussian_noise(float* __restrict__ x, long long n,
                                     float sigma, unsigned long long seed,
                                     unsigned long long offset) {
  long long n4 = (n + 3) >> 2;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n4;
       i += (long long)gridDim.x * blockDim.x) {
    hiprandStatePhilox4_32_10_t st;
    hiprand_init(seed, (unsigned long long)i, offset, &st);
    float4 r = hiprand_normal4(&st);
    long long base = i << 2;
    float rv[4] = {r.x, r.y, r.z, r.w};
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      if (base + j < n) x[base + j] = fmaf(sigma, rv[j], x[base + j]);
  }
}

// ---------------------------------------------------------------------------
// Fused optimizer steps (torch.optim semantics) on flat buffers
// ---------------------------------------------------------------------------

__global__ void k_sgd_step(float* __restrict__ p, const float* __restrict__ g,
                           float* __restrict__ buf, float lr,
                           const float* __restrict__ lr_ptr, float momentum,
                           float dampening, float weight_decay, int nesterov,
                           int first_step, long long n) {
  // lr from a device scalar when lr_ptr != null — lets a hipGraph replay
  // the step while the host retunes the LR between rounds
  if (lr_ptr) lr = *lr_ptr;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    float dp = g[i];
    if (weight_decay != 0.f) dp = fmaf(weight_decay, p[i], dp);
    if (momentum != 0.f) {
      float b = first_step ? dp : fmaf(momentum, buf[i], (1.f - dampening) * dp);
      buf[i] = b;
      dp = nesterov ? fmaf(momentum, b, dp) : b;
    }
    p[i] = fmaf(-lr, dp, p[i]);
  }
}

__global__ void k_adam_step(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            float* __restrict__ vmax, float lr, float beta1,
                            float beta2, float eps, float weight_decay,
                            float bc1, float bc2, int amsgrad, int adamw,
                            long long n) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    float gi = g[i];
    float pi = p[i];
    if (weight_decay != 0.f) {
      if (adamw) pi *= (1.f - lr * weight_decay);
      else gi = fmaf(weight_decay, pi, gi);
    }
    float mi = fmaf(beta1, m[i], (1.f - beta1) * gi);
    float vi = fmaf(beta2, v[i], (1.f - beta2) * gi * gi);
    m[i] = mi; v[i] = vi;
    float vhat;
    if (amsgrad) {
      float vm = fmaxf(vmax[i], vi);
      vmax[i] = vm;
      vhat = vm / bc2;
    } else {
      vhat = vi / bc2;
    }
    p[i] = pi - (lr / bc1) * mi / (sqrtf(vhat) + eps);
  }
}

__global__ void k_adamax_step(float* __restrict__ p, const float* __restrict__ g,
                              float* __restrict__ m, float* __restrict__ u,
                              float lr, float beta1, float beta2, float eps,
                              float weight_decay, float bc1, long long n) {
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    float gi = g[i];
    if (weight_decay != 0.f) gi = fmaf(weight_decay, p[i], gi);
    float mi = fmaf(beta1, m[i], (1.f - beta1) * gi);
    float ui = fmaxf(beta2 * u[i], fabsf(gi) + eps);
    m[i] = mi; u[i] = ui;
    p[i] = p[i] - (lr / bc1) * mi / ui;
  }
}

// ---------------------------------------------------------------------------
// Segmented Σx² (per-layer norms for LAMB/LARS trust ratios)
// ---------------------------------------------------------------------------

#define SEG_BLOCKS_PER 8

__global__ void k_segmented_sqnorm(const float* __restrict__ x,
                                   const long long* __restrict__ offs,
                                   int n_segs, double* __restrict__ out) {
  int seg = blockIdx.x / SEG_BLOCKS_PER;
  int sub = blockIdx.x % SEG_BLOCKS_PER;
  if (seg >= n_segs) return;
  long long lo = offs[seg], hi = offs[seg + 1];
  double q = 0.0;
  for (long long i = lo + sub * (long long)blockDim.x + threadIdx.x; i < hi;
       i += (long long)SEG_BLOCKS_PER * blockDim.x) {
    double v = x[i];
    q += v * v;
  }
  for (int d = WAVE / 2; d > 0; d >>= 1) q += __shfl_down(q, d, WAVE);
  __shared__ double lds[BLOCK / WAVE];
  int wave = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
  if (lane == 0) lds[wave] = q;
  __syncthreads();
  if (threadIdx.x == 0) {
    double t = 0;
    for (int w = 0; w < BLOCK / WAVE; ++w) t += lds[w];
    atomicAdd(&out[seg], t);
  }
}

// ---------------------------------------------------------------------------
// Quantization binning + sparsify (stats come in as device scalars so the
// whole pipeline stays on-device; semantics per reference quant.py:76-100)
// ---------------------------------------------------------------------------

__global__ void k_quant_bin_mask(float* __restrict__ x, long long n,
                                 const float* __restrict__ min_t,
                                 const float* __restrict__ max_t,
                                 const float* __restrict__ thresh_t,
                                 int n_bins) {
  float mn = *min_t, mx = *max_t, th = *thresh_t;
  float w = (n_bins > 1) ? (mx - mn) / (float)(n_bins - 1) : 0.f;
  float inv_w = (w != 0.f) ? 1.f / w : 0.f;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    float v = x[i];
    if (fabsf(v) <= th) { x[i] = 0.f; continue; }
    // bucketize(v - w/2, linspace(mn,mx,n_bins), right=False) == ceil((y-mn)/w)
    float y = v - 0.5f * w;
    int idx = (w != 0.f) ? (int)ceilf((y - mn) * inv_w) : 0;
    idx = idx < 0 ? 0 : (idx > n_bins - 1 ? n_bins - 1 : idx);
    x[i] = fmaf((float)idx, w, mn);
  }
}

// ---------------------------------------------------------------------------
// Fused GRU gate math (nlg_gru recurrence, eval/serving path): given the
// precomputed projections g_i = W_ih·x_t and g_h = W_hh·h (each [B, 3H],
// row-major) and h [B, H], computes
//   r = σ(i_r+h_r); z = σ(i_i+h_i); n = tanh(i_n + r·h_n); h' = n + z·(h−n)
// in ONE kernel instead of the reference's 6+ eager ops per step
// (experiments/nlg_gru/model.py:20-28).
// ---------------------------------------------------------------------------

__global__ void k_gru_gates(const float* __restrict__ g_i,
                            const float* __restrict__ g_h,
                            const float* __restrict__ h,
                            float* __restrict__ out, int B, int H) {
  long long n = (long long)B * H;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    long long row = i / H, col = i % H;
    long long base = row * 3LL * H + col;
    float r = 1.f / (1.f + __expf(-(g_i[base] + g_h[base])));
    float z = 1.f / (1.f + __expf(-(g_i[base + H] + g_h[base + H])));
    float nn = tanhf(g_i[base + 2LL * H] + r * g_h[base + 2LL * H]);
    out[i] = nn + z * (h[i] - nn);
  }
}

// ---------------------------------------------------------------------------
// C-linkage launchers (called from bindings.cpp on the torch stream)
// ---------------------------------------------------------------------------

extern "C" {

void launch_pseudo_grad(float* out, const float* ws, const float* wt,
                        float weight, long long n, hipStream_t s) {
  hipLaunchKernelGGL(k_pseudo_grad, dim3(grid_for(n >> 2)), dim3(BLOCK), 0, s,
                     out, ws, wt, weight, n);
}

void launch_axpy(float* y, const float* x, float alpha, long long n,
                 hipStream_t s) {
  hipLaunchKernelGGL(k_axpy, dim3(grid_for(n >> 2)), dim3(BLOCK), 0, s,
                     y, x, alpha, n);
}

void launch_scale(float* x, float alpha, long long n, hipStream_t s) {
  hipLaunchKernelGGL(k_scale, dim3(grid_for(n >> 2)), dim3(BLOCK), 0, s,
                     x, alpha, n);
}

void launch_sum_sumsq(const float* x, long long n, double* out2,
                      hipStream_t s) {
  hipLaunchKernelGGL(k_sum_sumsq, dim3(grid_for_reduce(n >> 2)), dim3(BLOCK),
                     0, s, x, n, out2);
}

// two-stage: partials buffer must hold 2*grid doubles; out2 zero-init
void launch_sum_sumsq2(const float* x, long long n, double* partials,
                       double* out2, hipStream_t s) {
  int grid = grid_for(n >> 2);
  hipLaunchKernelGGL(k_sum_sumsq_partial, dim3(grid), dim3(BLOCK), 0, s,
                     x, n, partials);
  hipLaunchKernelGGL(k_fold_partials, dim3(1), dim3(BLOCK), 0, s,
                     partials, grid, out2);
}

int sum_sumsq2_partials(long long n) { return grid_for(n >> 2); }

void launch_clip_apply(float* x, long long n, const double* out2,
                       float max_norm, float eps, float* norm_out,
                       hipStream_t s) {
  hipLaunchKernelGGL(k_clip_apply, dim3(grid_for(n >> 2)), dim3(BLOCK), 0, s,
                     x, n, out2, max_norm, eps, norm_out);
}

void launch_add_gaussian_noise(float* x, long long n, float sigma,
                               unsigned long long seed,
                               unsigned long long offset, hipStream_t s) {
  hipLaunchKernelGGL(k_add_gaussian_noise, dim3(grid_for((n + 3) >> 2)),
                     dim3(BLOCK), 0, s, x, n, sigma, seed, offset);
}

void launch_sgd_step(float* p, const float* g, float* buf, float lr,
                     const float* lr_ptr, float momentum, float dampening,
                     float weight_decay, int nesterov, int first_step,
                     long long n, hipStream_t s) {
  hipLaunchKernelGGL(k_sgd_step, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     p, g, buf, lr, lr_ptr, momentum, dampening, weight_decay,
                     nesterov, first_step, n);
}

void launch_clip_apply_stats(float* x, long long n, const double* out2,
                             float max_norm, float eps, float* stats_acc,
                             hipStream_t s) {
  hipLaunchKernelGGL(k_clip_apply_stats, dim3(grid_for(n >> 2)), dim3(BLOCK),
                     0, s, x, n, out2, max_norm, eps, stats_acc);
}

void launch_adam_step(float* p, const float* g, float* m, float* v,
                      float* vmax, float lr, float beta1, float beta2,
                      float eps, float weight_decay, float bc1, float bc2,
                      int amsgrad, int adamw, long long n, hipStream_t s) {
  hipLaunchKernelGGL(k_adam_step, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     p, g, m, v, vmax, lr, beta1, beta2, eps, weight_decay,
                     bc1, bc2, amsgrad, adamw, n);
}

void launch_adamax_step(float* p, const float* g, float* m, float* u,
                        float lr, float beta1, float beta2, float eps,
                        float weight_decay, float bc1, long long n,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_adamax_step, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     p, g, m, u, lr, beta1, beta2, eps, weight_decay, bc1, n);
}

void launch_segmented_sqnorm(const float* x, const long long* offs,
                             int n_segs, double* out, hipStream_t s) {
  hipLaunchKernelGGL(k_segmented_sqnorm, dim3(n_segs * SEG_BLOCKS_PER),
                     dim3(BLOCK), 0, s, x, offs, n_segs, out);
}

void launch_quant_bin_mask(float* x, long long n, const float* min_t,
                           const float* max_t, const float* thresh_t,
                           int n_bins, hipStream_t s) {
  hipLaunchKernelGGL(k_quant_bin_mask, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     x, n, min_t, max_t, thresh_t, n_bins);
}

void launch_gru_gates(const float* g_i, const float* g_h, const float* h,
                      float* out, int B, int H, hipStream_t s) {
  hipLaunchKernelGGL(k_gru_gates, dim3(grid_for((long long)B * H)),
                     dim3(BLOCK), 0, s, g_i, g_h, h, out, B, H);
}

}  // extern "C"
