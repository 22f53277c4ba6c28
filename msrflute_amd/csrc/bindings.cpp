// Python bindings for the gfx950 flat-arena kernels (flat_ops.hip).
//
// Host side is written against the native ROCm/HIP ATen surface of
// PyTorch-ROCm (c10::hip) — no CUDA compatibility layer.  All launches go
// onto the current torch stream; no host synchronization anywhere.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

extern "C" {
void launch_pseudo_grad(float*, const float*, const float*, float, long long,
                        hipStream_t);
void launch_axpy(float*, const float*, float, long long, hipStream_t);
void launch_scale(float*, float, long long, hipStream_t);
void launch_sum_sumsq(const float*, long long, double*, hipStream_t);
void launch_sum_sumsq2(const float*, long long, double*, double*, hipStream_t);
int sum_sumsq2_partials(long long);
void launch_clip_apply(float*, long long, const double*, float, float, float*,
                       hipStream_t);
void launch_add_gaussian_noise(float*, long long, float, unsigned long long,
                               unsigned long long, hipStream_t);
void launch_sgd_step(float*, const float*, float*, float, const float*, float,
                     float, float, int, int, long long, hipStream_t);
void launch_clip_apply_stats(float*, long long, const double*, float, float,
                             float*, hipStream_t);
void launch_adam_step(float*, const float*, float*, float*, float*, float,
                      float, float, float, float, float, float, int, int,
                      long long, hipStream_t);
void launch_adamax_step(float*, const float*, float*, float*, float, float,
                        float, float, float, float, long long, hipStream_t);
void launch_segmented_sqnorm(const float*, const long long*, int, double*,
                             hipStream_t);
void launch_quant_bin_mask(float*, long long, const float*, const float*,
                           const float*, int, hipStream_t);
void launch_gru_gates(const float*, const float*, const float*, float*, int,
                      int, hipStream_t);
}

struct CnnWorkspace {
  float *xb, *a1, *r2, *a2, *z3, *a3, *dlogits, *dz3, *da2, *dz2, *dz1;
  float *wsl;  // split-K partial slab (B*18432 floats) — keep in sync
               // with the definition in fused_cnn.hip
  float *w2t, *w2rot;  // per-batch W2 fragment layouts (18432 floats each)
  int *yb;
  unsigned char *pidx, *m2, *m3;
  double *red_partials, *red_acc;
};

extern "C" void launch_lstm_seq_fwd(const float*, const float*, float*,
                                    float*, float*, int, int, int,
                                    hipStream_t);
extern "C" void launch_lstm_seq_bwd(const float*, const float*, const float*,
                                    const float*, float*, int, int, int,
                                    hipStream_t);
extern "C" void launch_lstm_seq_fwd_b(const float*, const float*, float*,
                                      float*, float*, int, int, int,
                                      hipStream_t);
extern "C" void launch_lstm_seq_bwd_b(const float*, const float*,
                                      const float*, const float*, float*,
                                      int, int, int, hipStream_t);
extern "C" void launch_gru_seq_fwd(const float*, const float*, const float*,
                                   float*, float*, float*, int, int,
                                   hipStream_t);
extern "C" void launch_gru_seq_bwd(const float*, const float*, const float*,
                                   const float*, const float*, float*,
                                   float*, int, int, hipStream_t);
extern "C" void launch_cnn_epoch(
    const float* shard_x, const long long* shard_y, const long long* order,
    long long n, int bs, int C, float* params, float* grads,
    CnnWorkspace ws, const float* lr_t, float max_norm, float p1, float p2,
    float* stats_acc, float* loss_acc, unsigned long long seed,
    hipStream_t s, long long row_base, int use_bf16);
extern "C" void launch_w2_layouts(const float*, float*, float*, hipStream_t);
extern "C" void launch_conv2_fwd_mfma(const float*, const float*,
                                      const float*, int, float*, hipStream_t);
extern "C" void launch_conv2_bwd_x_mfma(const float*, const float*,
                                        const float*, int, float*,
                                        hipStream_t);
extern "C" void launch_conv2_bwd_w_mfma(const float*, const float*, int,
                                        float*, float*, float*, float*,
                                        hipStream_t);
extern "C" void launch_fc1_fwd_mfma(const float*, const float*, const float*,
                                    int, float, unsigned long long,
                                    unsigned long long, float*, float*,
                                    float*, unsigned char*, hipStream_t);
extern "C" void launch_fc1_bwd_w_mfma(const float*, const float*, int,
                                      float*, float*, hipStream_t);
extern "C" void launch_fc1_bwd_x_mfma(const float*, const float*, int,
                                      float*, hipStream_t);
extern "C" void launch_mfma_bf16_probe(const void*, const void*, float*,
                                       hipStream_t);
extern "C" void launch_fc1_fwd_mfma_bf16(const float*, const float*,
                                         const float*, int, float,
                                         unsigned long long,
                                         unsigned long long, float*, float*,
                                         float*, unsigned char*, hipStream_t);
extern "C" void launch_conv2_fwd_mfma_bf16(const float*, const float*,
                                           const float*, int, float*,
                                           hipStream_t);
extern "C" void launch_conv2_bwd_x_mfma_bf16(const float*, const float*,
                                             const float*, int, float*,
                                             hipStream_t);
extern "C" void launch_conv2_bwd_w_mfma_bf16(const float*, const float*, int,
                                             float*, float*, float*,
                                             hipStream_t);

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
    float* stats_out, float* loss_out, hipStream_t s, int use_bf16);

extern "C" void launch_mega_clip_sgd(float*, float*, long long, int,
                                     double*, float, const float*, float*,
                                     hipStream_t);
extern "C" void launch_mega_pseudo_accum(float*, const float*, const float*,
                                         const float*, float*, long long,
                                         int, hipStream_t);

extern "C" void launch_cnn_round(
    const float* shard_x, const long long* shard_y, const long long* orders,
    const long long* row_bases, const long long* order_offs,
    const long long* counts, const float* weights,
    const unsigned long long* seeds, int K, int bs, int C,
    const float* server_params, float* params, float* grads,
    float* round_accum, CnnWorkspace ws, const float* lr_t, float max_norm,
    float p1, float p2, float* stats_out, float* loss_out, hipStream_t s,
    int use_bf16);

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_flat(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

void pseudo_grad(torch::Tensor out, torch::Tensor ws, torch::Tensor wt,
                 double weight) {
  check_flat(out, "out"); check_flat(ws, "ws"); check_flat(wt, "wt");
  TORCH_CHECK(out.numel() == ws.numel() && ws.numel() == wt.numel());
  launch_pseudo_grad(out.data_ptr<float>(), ws.data_ptr<float>(),
                     wt.data_ptr<float>(), (float)weight, out.numel(),
                     cur_stream());
}

void axpy(torch::Tensor y, torch::Tensor x, double alpha) {
  check_flat(y, "y"); check_flat(x, "x");
  TORCH_CHECK(y.numel() == x.numel());
  launch_axpy(y.data_ptr<float>(), x.data_ptr<float>(), (float)alpha,
              y.numel(), cur_stream());
}

void scale(torch::Tensor x, double alpha) {
  check_flat(x, "x");
  launch_scale(x.data_ptr<float>(), (float)alpha, x.numel(), cur_stream());
}

// two-stage reduction helper: no f64 atomics on the hot path
static torch::Tensor sum_sumsq_acc(const torch::Tensor& x) {
  auto acc = torch::zeros({2}, x.options().dtype(torch::kFloat64));
  int nparts = sum_sumsq2_partials(x.numel());
  auto partials = torch::empty({2 * (long long)nparts},
                               x.options().dtype(torch::kFloat64));
  launch_sum_sumsq2(x.data_ptr<float>(), x.numel(),
                    partials.data_ptr<double>(), acc.data_ptr<double>(),
                    cur_stream());
  return acc;
}

torch::Tensor sum_sumsq(torch::Tensor x) {
  check_flat(x, "x");
  return sum_sumsq_acc(x).to(torch::kFloat32);
}

torch::Tensor clip_by_norm(torch::Tensor x, double max_norm, double eps) {
  check_flat(x, "x");
  auto acc = sum_sumsq_acc(x);
  auto norm = torch::empty({}, x.options());
  launch_clip_apply(x.data_ptr<float>(), x.numel(), acc.data_ptr<double>(),
                    (float)max_norm, (float)eps, norm.data_ptr<float>(),
                    cur_stream());
  return norm;
}

void add_gaussian_noise(torch::Tensor x, double sigma, int64_t seed,
                        int64_t offset) {
  check_flat(x, "x");
  launch_add_gaussian_noise(x.data_ptr<float>(), x.numel(), (float)sigma,
                            (unsigned long long)seed,
                            (unsigned long long)offset, cur_stream());
}

void sgd_step(torch::Tensor p, torch::Tensor g, torch::Tensor buf, double lr,
              double momentum, double dampening, double weight_decay,
              bool nesterov, bool first_step) {
  check_flat(p, "p"); check_flat(g, "g");
  if (momentum != 0.0) {
    check_flat(buf, "buf");
    TORCH_CHECK(buf.numel() == p.numel(), "momentum buffer size mismatch");
  }
  launch_sgd_step(p.data_ptr<float>(), g.data_ptr<float>(),
                  buf.numel() ? buf.data_ptr<float>() : nullptr, (float)lr,
                  nullptr, (float)momentum, (float)dampening,
                  (float)weight_decay, nesterov ? 1 : 0, first_step ? 1 : 0,
                  p.numel(), cur_stream());
}

// graph-replayable SGD: lr read from a 1-element device tensor
void sgd_step_devlr(torch::Tensor p, torch::Tensor g, torch::Tensor buf,
                    torch::Tensor lr_t, double momentum, double dampening,
                    double weight_decay, bool nesterov, bool first_step) {
  check_flat(p, "p"); check_flat(g, "g"); check_flat(lr_t, "lr_t");
  if (momentum != 0.0) {
    check_flat(buf, "buf");
    TORCH_CHECK(buf.numel() == p.numel(), "momentum buffer size mismatch");
  }
  launch_sgd_step(p.data_ptr<float>(), g.data_ptr<float>(),
                  buf.numel() ? buf.data_ptr<float>() : nullptr, 0.f,
                  lr_t.data_ptr<float>(), (float)momentum, (float)dampening,
                  (float)weight_decay, nesterov ? 1 : 0, first_step ? 1 : 0,
                  p.numel(), cur_stream());
}

// fused per-batch clip + sufficient-stats accumulate (one reduction pass):
// clips x to max_norm and adds post-clip {Σx, Σx²} into stats_acc[0..1]
void clip_stats_accumulate(torch::Tensor x, double max_norm, double eps,
                           torch::Tensor stats_acc) {
  check_flat(x, "x"); check_flat(stats_acc, "stats_acc");
  TORCH_CHECK(stats_acc.numel() >= 2, "stats_acc must have 2 elements");
  auto acc = sum_sumsq_acc(x);
  launch_clip_apply_stats(x.data_ptr<float>(), x.numel(),
                          acc.data_ptr<double>(), (float)max_norm, (float)eps,
                          stats_acc.data_ptr<float>(), cur_stream());
}

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor vmax, int64_t step, double lr,
               double beta1, double beta2, double eps, double weight_decay,
               bool amsgrad, bool adamw) {
  check_flat(p, "p"); check_flat(g, "g"); check_flat(m, "m"); check_flat(v, "v");
  if (amsgrad) check_flat(vmax, "vmax");
  float bc1 = 1.0f - (float)std::pow(beta1, (double)step);
  float bc2 = 1.0f - (float)std::pow(beta2, (double)step);
  launch_adam_step(p.data_ptr<float>(), g.data_ptr<float>(),
                   m.data_ptr<float>(), v.data_ptr<float>(),
                   vmax.numel() ? vmax.data_ptr<float>() : nullptr, (float)lr,
                   (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                   bc1, bc2, amsgrad ? 1 : 0, adamw ? 1 : 0, p.numel(),
                   cur_stream());
}

void adamax_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor u, int64_t step, double lr, double beta1,
                 double beta2, double eps, double weight_decay) {
  check_flat(p, "p"); check_flat(g, "g"); check_flat(m, "m"); check_flat(u, "u");
  float bc1 = 1.0f - (float)std::pow(beta1, (double)step);
  launch_adamax_step(p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), u.data_ptr<float>(), (float)lr,
                     (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bc1, p.numel(), cur_stream());
}

torch::Tensor segmented_sqnorm(torch::Tensor x, torch::Tensor seg_offsets) {
  check_flat(x, "x");
  TORCH_CHECK(seg_offsets.is_cuda() && seg_offsets.is_contiguous() &&
              seg_offsets.scalar_type() == torch::kInt64,
              "seg_offsets must be contiguous int64 on GPU");
  int n_segs = (int)seg_offsets.numel() - 1;
  auto out = torch::zeros({n_segs}, x.options().dtype(torch::kFloat64));
  launch_segmented_sqnorm(
      x.data_ptr<float>(),
      reinterpret_cast<const long long*>(seg_offsets.data_ptr<int64_t>()),
      n_segs, out.data_ptr<double>(), cur_stream());
  return out.to(torch::kFloat32);
}

void quant_bin_mask(torch::Tensor x, torch::Tensor min_t, torch::Tensor max_t,
                    torch::Tensor thresh_t, int64_t n_bins) {
  check_flat(x, "x");
  check_flat(min_t, "min_t"); check_flat(max_t, "max_t");
  check_flat(thresh_t, "thresh_t");
  launch_quant_bin_mask(x.data_ptr<float>(), x.numel(),
                        min_t.data_ptr<float>(), max_t.data_ptr<float>(),
                        thresh_t.data_ptr<float>(), (int)n_bins, cur_stream());
}

// fully-fused CNN-FEMNIST client epoch (fused_cnn.hip): one host call
// trains a whole local epoch — fwd, bwd, clip+stats, SGD per batch —
// bypassing autograd and hip graphs entirely.
static CnnWorkspace slice_ws(torch::Tensor&, torch::Tensor&,
                             torch::Tensor&, torch::Tensor&, int, int);

void cnn_epoch(torch::Tensor shard_x, torch::Tensor shard_y,
               torch::Tensor order, int64_t bs, int64_t C,
               torch::Tensor params, torch::Tensor grads,
               torch::Tensor work_f, torch::Tensor work_i,
               torch::Tensor work_b, torch::Tensor work_d,
               torch::Tensor lr_t, double max_norm, double p1, double p2,
               torch::Tensor stats_acc, torch::Tensor loss_acc,
               int64_t seed, bool use_bf16) {
  check_flat(params, "params"); check_flat(grads, "grads");
  check_flat(lr_t, "lr_t"); check_flat(stats_acc, "stats_acc");
  check_flat(loss_acc, "loss_acc"); check_flat(work_f, "work_f");
  TORCH_CHECK(shard_x.is_cuda() && shard_x.is_contiguous() &&
              shard_x.scalar_type() == torch::kFloat32, "shard_x");
  TORCH_CHECK(shard_y.is_cuda() && shard_y.is_contiguous() &&
              shard_y.scalar_type() == torch::kInt64, "shard_y");
  TORCH_CHECK(order.is_cuda() && order.scalar_type() == torch::kInt64,
              "order must be int64 on device");
  long long n = shard_y.numel();
  TORCH_CHECK(shard_x.numel() == n * 784, "shard_x must be [n,784]");
  TORCH_CHECK(bs >= 1 && bs <= 32, "fused CNN epoch supports batch <= 32");
  int B = (int)bs;
  // ONE shared slicer (slice_ws) lays out the workspace — a duplicated
  // inline copy here once drifted from it and left ws.wsl dangling
  CnnWorkspace ws = slice_ws(work_f, work_i, work_b, work_d, B, (int)C);
  launch_cnn_epoch(shard_x.data_ptr<float>(),
                   reinterpret_cast<const long long*>(
                       shard_y.data_ptr<int64_t>()),
                   reinterpret_cast<const long long*>(
                       order.data_ptr<int64_t>()),
                   n, B, (int)C,
                   params.data_ptr<float>(), grads.data_ptr<float>(), ws,
                   lr_t.data_ptr<float>(), (float)max_norm, (float)p1,
                   (float)p2, stats_acc.data_ptr<float>(),
                   loss_acc.data_ptr<float>(), (unsigned long long)seed,
                   cur_stream(), 0, use_bf16 ? 1 : 0);
}

static CnnWorkspace slice_ws(torch::Tensor& work_f, torch::Tensor& work_i,
                             torch::Tensor& work_b, torch::Tensor& work_d,
                             int B, int C) {
  float* f = work_f.data_ptr<float>();
  CnnWorkspace ws;
  ws.xb = f;            f += (long long)B * 784;
  ws.a1 = f;            f += (long long)B * 21632;
  ws.r2 = f;            f += (long long)B * 36864;
  ws.a2 = f;            f += (long long)B * 9216;
  ws.z3 = f;            f += (long long)B * 128;
  ws.a3 = f;            f += (long long)B * 128;
  ws.dlogits = f;       f += (long long)B * C;
  ws.dz3 = f;           f += (long long)B * 128;
  ws.da2 = f;           f += (long long)B * 9216;
  ws.dz2 = f;           f += (long long)B * 36864;
  ws.dz1 = f;           f += (long long)B * 21632;
  ws.wsl = f;           f += (long long)B * 18432;  // split-K partial slab
  ws.w2t = f;           f += 18432;
  ws.w2rot = f;         f += 18432;
  TORCH_CHECK(f - work_f.data_ptr<float>() <= work_f.numel(),
              "float workspace too small");
  TORCH_CHECK(work_i.numel() >= B, "int workspace too small");
  TORCH_CHECK(work_d.numel() >= 2 * 2048 + 2, "double workspace too small");
  ws.yb = work_i.data_ptr<int>();
  unsigned char* u = work_b.data_ptr<unsigned char>();
  ws.pidx = u;          u += (long long)B * 9216;
  ws.m2 = u;            u += (long long)B * 9216;
  ws.m3 = u;            u += (long long)B * 128;
  TORCH_CHECK(u - work_b.data_ptr<unsigned char>() <= work_b.numel(),
              "byte workspace too small");
  ws.red_partials = work_d.data_ptr<double>();
  ws.red_acc = work_d.data_ptr<double>() + work_d.numel() - 2;
  return ws;
}

// whole-round driver: K clients in ONE host call (copy-in, fused epoch,
// weighted pseudo-gradient, accumulate) — per-client loss/stats in slots
void cnn_round(torch::Tensor shard_x, torch::Tensor shard_y,
               torch::Tensor orders_dev, torch::Tensor row_bases,
               torch::Tensor order_offs, torch::Tensor counts,
               torch::Tensor weights, torch::Tensor seeds, int64_t bs,
               int64_t C, torch::Tensor server_params, torch::Tensor params,
               torch::Tensor grads, torch::Tensor round_accum,
               torch::Tensor work_f, torch::Tensor work_i,
               torch::Tensor work_b, torch::Tensor work_d,
               torch::Tensor lr_t, double max_norm, double p1, double p2,
               torch::Tensor stats_out, torch::Tensor loss_out,
               bool use_bf16) {
  check_flat(server_params, "server_params"); check_flat(params, "params");
  check_flat(grads, "grads"); check_flat(round_accum, "round_accum");
  TORCH_CHECK(orders_dev.is_cuda() && orders_dev.scalar_type() == torch::kInt64);
  TORCH_CHECK(!row_bases.is_cuda() && !order_offs.is_cuda() &&
              !counts.is_cuda() && !weights.is_cuda() && !seeds.is_cuda(),
              "per-client metadata must be host tensors");
  TORCH_CHECK(row_bases.scalar_type() == torch::kInt64 &&
              order_offs.scalar_type() == torch::kInt64 &&
              counts.scalar_type() == torch::kInt64 &&
              seeds.scalar_type() == torch::kInt64 &&
              weights.scalar_type() == torch::kFloat32,
              "metadata dtypes: int64 (weights fp32)");
  TORCH_CHECK(row_bases.is_contiguous() && order_offs.is_contiguous() &&
              counts.is_contiguous() && weights.is_contiguous() &&
              seeds.is_contiguous());
  int K = (int)counts.numel();
  TORCH_CHECK(stats_out.numel() >= 2 * K && loss_out.numel() >= K);
  TORCH_CHECK(bs >= 1 && bs <= 32);
  TORCH_CHECK(row_bases.numel() == K && order_offs.numel() == K &&
              weights.numel() == K && seeds.numel() == K,
              "per-client metadata lengths must all equal K");
  // bound every per-client (offset, count) against the shard / order
  // tensors so inconsistent host metadata raises instead of launching
  // out-of-bounds device reads
  {
    long long n_rows = shard_y.numel();
    TORCH_CHECK(shard_x.numel() == n_rows * 784,
                "shard_x rows must match shard_y (784 features each)");
    long long n_orders = orders_dev.numel();
    auto rb = row_bases.accessor<int64_t, 1>();
    auto oo = order_offs.accessor<int64_t, 1>();
    auto ct = counts.accessor<int64_t, 1>();
    for (int k = 0; k < K; ++k) {
      TORCH_CHECK(ct[k] >= 0 && rb[k] >= 0 && oo[k] >= 0,
                  "negative per-client metadata at k=", k);
      TORCH_CHECK(rb[k] + ct[k] <= n_rows,
                  "client ", k, ": row_base+count ", rb[k] + ct[k],
                  " exceeds shard rows ", n_rows);
      TORCH_CHECK(oo[k] + ct[k] <= n_orders,
                  "client ", k, ": order_off+count ", oo[k] + ct[k],
                  " exceeds orders length ", n_orders);
    }
  }
  CnnWorkspace ws = slice_ws(work_f, work_i, work_b, work_d, (int)bs, (int)C);
  launch_cnn_round(
      shard_x.data_ptr<float>(),
      reinterpret_cast<const long long*>(shard_y.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(orders_dev.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(row_bases.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(order_offs.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(counts.data_ptr<int64_t>()),
      weights.data_ptr<float>(),
      reinterpret_cast<const unsigned long long*>(seeds.data_ptr<int64_t>()),
      K, (int)bs, (int)C, server_params.data_ptr<float>(),
      params.data_ptr<float>(), grads.data_ptr<float>(),
      round_accum.data_ptr<float>(), ws, lr_t.data_ptr<float>(),
      (float)max_norm, (float)p1, (float)p2, stats_out.data_ptr<float>(),
      loss_out.data_ptr<float>(), cur_stream(), use_bf16 ? 1 : 0);
}

// fused LSTM sequence recurrence (lstm_seq.hip): forward over all T
// steps in one launch; returns (h_seq, activated gates, cell states)
std::vector<torch::Tensor> lstm_seq_fwd(torch::Tensor xp,
                                        torch::Tensor w_hh) {
  check_flat(xp, "xp"); check_flat(w_hh, "w_hh");
  TORCH_CHECK(xp.dim() == 3 && xp.size(2) == 4 * 256 &&
              w_hh.numel() == 256 * 4 * 256,
              "fused LSTM supports hidden size 256; pass W_hh "
              "float4-packed [H/4, 4H, 4] (ops/lstm._pack_fwd)");
  long long B = xp.size(0), T = xp.size(1);
  auto h_seq = torch::empty({B, T, 256}, xp.options());
  auto gates = torch::empty({B, T, 4 * 256}, xp.options());
  auto c_seq = torch::empty({B, T, 256}, xp.options());
  launch_lstm_seq_fwd(xp.data_ptr<float>(), w_hh.data_ptr<float>(),
                      h_seq.data_ptr<float>(), gates.data_ptr<float>(),
                      c_seq.data_ptr<float>(), (int)B, (int)T, 256,
                      cur_stream());
  return {h_seq, gates, c_seq};
}

torch::Tensor lstm_seq_bwd(torch::Tensor gates, torch::Tensor c_seq,
                           torch::Tensor w_hh, torch::Tensor dh_out) {
  check_flat(gates, "gates"); check_flat(c_seq, "c_seq");
  check_flat(w_hh, "w_hh"); check_flat(dh_out, "dh_out");
  TORCH_CHECK(w_hh.numel() == 4 * 256 * 256,
              "lstm_seq_bwd wants W_hh float4-packed [4H/4, H, 4] "
              "(ops/lstm._pack_bwd)");
  long long B = gates.size(0), T = gates.size(1);
  auto dg = torch::empty_like(gates);
  launch_lstm_seq_bwd(gates.data_ptr<float>(), c_seq.data_ptr<float>(),
                      w_hh.data_ptr<float>(), dh_out.data_ptr<float>(),
                      dg.data_ptr<float>(), (int)B, (int)T, 256,
                      cur_stream());
  return dg;
}

// batched (cross-client) LSTM recurrence: xp [R, T, 4H], K-stacked
// transposed hidden weights [K, H, 4H]; row r belongs to client
// r / rows_per_client (Shakespeare mega round)
std::vector<torch::Tensor> lstm_seq_fwd_b(torch::Tensor xp,
                                          torch::Tensor w_hh_t_stack,
                                          int64_t rows_per_client) {
  check_flat(xp, "xp"); check_flat(w_hh_t_stack, "w_hh_t_stack");
  TORCH_CHECK(xp.dim() == 3 && xp.size(2) == 4 * 256,
              "xp must be [R, T, 4*256]");
  long long R = xp.size(0), T = xp.size(1);
  TORCH_CHECK(rows_per_client > 0 && R % rows_per_client == 0);
  long long K = R / rows_per_client;
  TORCH_CHECK(w_hh_t_stack.numel() == K * 256 * 4 * 256,
              "w_hh_t_stack must be [K, 256, 4*256]");
  auto h_seq = torch::empty({R, T, 256}, xp.options());
  auto gates = torch::empty({R, T, 4 * 256}, xp.options());
  auto c_seq = torch::empty({R, T, 256}, xp.options());
  launch_lstm_seq_fwd_b(xp.data_ptr<float>(),
                        w_hh_t_stack.data_ptr<float>(),
                        h_seq.data_ptr<float>(), gates.data_ptr<float>(),
                        c_seq.data_ptr<float>(), (int)R,
                        (int)rows_per_client, (int)T, cur_stream());
  return {h_seq, gates, c_seq};
}

torch::Tensor lstm_seq_bwd_b(torch::Tensor gates, torch::Tensor c_seq,
                             torch::Tensor w_hh_stack,
                             torch::Tensor dh_out,
                             int64_t rows_per_client) {
  check_flat(gates, "gates"); check_flat(c_seq, "c_seq");
  check_flat(w_hh_stack, "w_hh_stack"); check_flat(dh_out, "dh_out");
  long long R = gates.size(0), T = gates.size(1);
  TORCH_CHECK(rows_per_client > 0 && R % rows_per_client == 0);
  long long K = R / rows_per_client;
  TORCH_CHECK(w_hh_stack.numel() == K * 4 * 256 * 256,
              "w_hh_stack must be [K, 4*256, 256]");
  auto dg = torch::empty_like(gates);
  launch_lstm_seq_bwd_b(gates.data_ptr<float>(), c_seq.data_ptr<float>(),
                        w_hh_stack.data_ptr<float>(),
                        dh_out.data_ptr<float>(), dg.data_ptr<float>(),
                        (int)R, (int)rows_per_client, (int)T,
                        cur_stream());
  return dg;
}

// fused GRU sequence recurrence (gru_seq.hip)
std::vector<torch::Tensor> gru_seq_fwd(torch::Tensor gi, torch::Tensor whh_t,
                                       torch::Tensor b_hh) {
  check_flat(gi, "gi"); check_flat(whh_t, "whh_t"); check_flat(b_hh, "b_hh");
  TORCH_CHECK(gi.dim() == 3 && gi.size(2) == 3 * 512 &&
              whh_t.numel() == 512 * 3 * 512,
              "fused GRU supports hidden size 512; pass W_hh "
              "float4-packed [H/4, 3H, 4] (ops/lstm._GRUSeq)");
  long long B = gi.size(0), T = gi.size(1);
  auto h_seq = torch::empty({B, T, 512}, gi.options());
  auto gates = torch::empty({B, T, 3 * 512}, gi.options());
  auto ghn = torch::empty({B, T, 512}, gi.options());
  launch_gru_seq_fwd(gi.data_ptr<float>(), whh_t.data_ptr<float>(),
                     b_hh.data_ptr<float>(), h_seq.data_ptr<float>(),
                     gates.data_ptr<float>(), ghn.data_ptr<float>(),
                     (int)B, (int)T, cur_stream());
  return {h_seq, gates, ghn};
}

std::vector<torch::Tensor> gru_seq_bwd(torch::Tensor gates, torch::Tensor ghn,
                                       torch::Tensor h_seq,
                                       torch::Tensor w_hh,
                                       torch::Tensor dh_out) {
  check_flat(gates, "gates"); check_flat(ghn, "ghn");
  check_flat(h_seq, "h_seq"); check_flat(w_hh, "w_hh");
  check_flat(dh_out, "dh_out");
  TORCH_CHECK(w_hh.numel() == 3 * 512 * 512,
              "gru_seq_bwd wants W_hh float4-packed [3H/4, H, 4]");
  long long B = gates.size(0), T = gates.size(1);
  auto dgi = torch::empty_like(gates);
  auto dgh = torch::empty_like(gates);
  launch_gru_seq_bwd(gates.data_ptr<float>(), ghn.data_ptr<float>(),
                     h_seq.data_ptr<float>(), w_hh.data_ptr<float>(),
                     dh_out.data_ptr<float>(), dgi.data_ptr<float>(),
                     dgh.data_ptr<float>(), (int)B, (int)T, cur_stream());
  return {dgi, dgh};
}

// fused GRU gate math (no-grad eval path of the nlg_gru recurrence)
torch::Tensor gru_gates(torch::Tensor g_i, torch::Tensor g_h,
                        torch::Tensor h) {
  check_flat(g_i, "g_i"); check_flat(g_h, "g_h"); check_flat(h, "h");
  int B = (int)h.size(0), H = (int)h.size(1);
  TORCH_CHECK(g_i.numel() == 3LL * B * H && g_h.numel() == 3LL * B * H,
              "g_i/g_h must be [B, 3H]");
  auto out = torch::empty_like(h);
  launch_gru_gates(g_i.data_ptr<float>(), g_h.data_ptr<float>(),
                   h.data_ptr<float>(), out.data_ptr<float>(), B, H,
                   cur_stream());
  return out;
}

}  // namespace

// ---------------------------------------------------------------------------
// per-kernel MFMA debug entries (numerics tests vs torch references;
// the epoch driver calls the same kernels internally)
// ---------------------------------------------------------------------------
torch::Tensor dbg_conv2_fwd_mfma(torch::Tensor a1, torch::Tensor w2,
                                 torch::Tensor b2, int64_t B) {
  check_flat(a1, "a1"); check_flat(w2, "w2"); check_flat(b2, "b2");
  TORCH_CHECK(a1.numel() >= B * 21632 && w2.numel() == 18432 &&
              b2.numel() == 64);
  auto r2 = torch::empty({B * 36864}, a1.options());
  auto w2t = torch::empty({18432}, a1.options());
  auto w2rot = torch::empty({18432}, a1.options());
  launch_w2_layouts(w2.data_ptr<float>(), w2t.data_ptr<float>(),
                    w2rot.data_ptr<float>(), cur_stream());
  launch_conv2_fwd_mfma(a1.data_ptr<float>(), w2t.data_ptr<float>(),
                        b2.data_ptr<float>(), (int)B, r2.data_ptr<float>(),
                        cur_stream());
  return r2;
}

torch::Tensor dbg_conv2_fwd_mfma_bf16(torch::Tensor a1, torch::Tensor w2,
                                      torch::Tensor b2, int64_t B) {
  check_flat(a1, "a1"); check_flat(w2, "w2"); check_flat(b2, "b2");
  auto r2 = torch::empty({B * 36864}, a1.options());
  launch_conv2_fwd_mfma_bf16(a1.data_ptr<float>(), w2.data_ptr<float>(),
                             b2.data_ptr<float>(), (int)B,
                             r2.data_ptr<float>(), cur_stream());
  return r2;
}

torch::Tensor dbg_conv2_bwd_x_mfma_bf16(torch::Tensor dz2, torch::Tensor w2,
                                        torch::Tensor a1, int64_t B) {
  check_flat(dz2, "dz2"); check_flat(w2, "w2"); check_flat(a1, "a1");
  auto dz1 = torch::empty({B * 21632}, dz2.options());
  launch_conv2_bwd_x_mfma_bf16(dz2.data_ptr<float>(), w2.data_ptr<float>(),
                               a1.data_ptr<float>(), (int)B,
                               dz1.data_ptr<float>(), cur_stream());
  return dz1;
}

std::vector<torch::Tensor> dbg_conv2_bwd_w_mfma_bf16(torch::Tensor dz2,
                                                     torch::Tensor a1,
                                                     int64_t B) {
  check_flat(dz2, "dz2"); check_flat(a1, "a1");
  auto slab = torch::empty({B * 18432}, dz2.options());
  auto dw2 = torch::empty({18432}, dz2.options());
  auto db2 = torch::empty({64}, dz2.options());
  launch_conv2_bwd_w_mfma_bf16(dz2.data_ptr<float>(), a1.data_ptr<float>(),
                               (int)B, slab.data_ptr<float>(),
                               dw2.data_ptr<float>(), db2.data_ptr<float>(),
                               cur_stream());
  return {dw2, db2};
}

std::vector<torch::Tensor> dbg_fc1_fwd_mfma_bf16(torch::Tensor a2,
                                                 torch::Tensor w3,
                                                 torch::Tensor b3, int64_t B,
                                                 double p2, int64_t seed,
                                                 int64_t offset) {
  check_flat(a2, "a2"); check_flat(w3, "w3"); check_flat(b3, "b3");
  auto slab = torch::empty({36 * B * 128}, a2.options());
  auto z3 = torch::empty({B * 128}, a2.options());
  auto a3 = torch::empty({B * 128}, a2.options());
  auto m3 = torch::empty({B * 128}, a2.options().dtype(torch::kUInt8));
  launch_fc1_fwd_mfma_bf16(a2.data_ptr<float>(), w3.data_ptr<float>(),
                           b3.data_ptr<float>(), (int)B, (float)p2,
                           (unsigned long long)seed,
                           (unsigned long long)offset,
                           slab.data_ptr<float>(), z3.data_ptr<float>(),
                           a3.data_ptr<float>(), m3.data_ptr<unsigned char>(),
                           cur_stream());
  return {z3, a3, m3};
}

torch::Tensor dbg_conv2_bwd_x_mfma(torch::Tensor dz2, torch::Tensor w2,
                                   torch::Tensor a1, int64_t B) {
  check_flat(dz2, "dz2"); check_flat(w2, "w2"); check_flat(a1, "a1");
  TORCH_CHECK(dz2.numel() >= B * 36864 && a1.numel() >= B * 21632);
  auto dz1 = torch::empty({B * 21632}, dz2.options());
  auto w2t = torch::empty({18432}, dz2.options());
  auto w2rot = torch::empty({18432}, dz2.options());
  launch_w2_layouts(w2.data_ptr<float>(), w2t.data_ptr<float>(),
                    w2rot.data_ptr<float>(), cur_stream());
  launch_conv2_bwd_x_mfma(dz2.data_ptr<float>(), w2rot.data_ptr<float>(),
                          a1.data_ptr<float>(), (int)B,
                          dz1.data_ptr<float>(), cur_stream());
  return dz1;
}

std::vector<torch::Tensor> dbg_conv2_bwd_w_mfma(torch::Tensor dz2,
                                                torch::Tensor a1, int64_t B) {
  check_flat(dz2, "dz2"); check_flat(a1, "a1");
  auto slab = torch::empty({B * 18432}, dz2.options());
  auto dz2t = torch::empty({B * 36864}, dz2.options());
  auto dw2 = torch::empty({18432}, dz2.options());
  auto db2 = torch::empty({64}, dz2.options());
  launch_conv2_bwd_w_mfma(dz2.data_ptr<float>(), a1.data_ptr<float>(),
                          (int)B, dz2t.data_ptr<float>(),
                          slab.data_ptr<float>(),
                          dw2.data_ptr<float>(), db2.data_ptr<float>(),
                          cur_stream());
  return {dw2, db2};
}

std::vector<torch::Tensor> dbg_fc1_fwd_mfma(torch::Tensor a2,
                                            torch::Tensor w3,
                                            torch::Tensor b3, int64_t B,
                                            double p2, int64_t seed,
                                            int64_t offset) {
  check_flat(a2, "a2"); check_flat(w3, "w3"); check_flat(b3, "b3");
  TORCH_CHECK(a2.numel() >= B * 9216 && w3.numel() == 128 * 9216);
  auto slab = torch::empty({64 * B * 128}, a2.options());
  auto z3 = torch::empty({B * 128}, a2.options());
  auto a3 = torch::empty({B * 128}, a2.options());
  auto m3 = torch::empty({B * 128}, a2.options().dtype(torch::kUInt8));
  launch_fc1_fwd_mfma(a2.data_ptr<float>(), w3.data_ptr<float>(),
                      b3.data_ptr<float>(), (int)B, (float)p2,
                      (unsigned long long)seed, (unsigned long long)offset,
                      slab.data_ptr<float>(), z3.data_ptr<float>(),
                      a3.data_ptr<float>(), m3.data_ptr<unsigned char>(),
                      cur_stream());
  return {z3, a3, m3};
}

std::vector<torch::Tensor> dbg_fc1_bwd_w_mfma(torch::Tensor dz3,
                                              torch::Tensor a2, int64_t B) {
  check_flat(dz3, "dz3"); check_flat(a2, "a2");
  auto dw3 = torch::empty({128 * 9216}, dz3.options());
  auto db3 = torch::empty({128}, dz3.options());
  launch_fc1_bwd_w_mfma(dz3.data_ptr<float>(), a2.data_ptr<float>(), (int)B,
                        dw3.data_ptr<float>(), db3.data_ptr<float>(),
                        cur_stream());
  return {dw3, db3};
}

torch::Tensor dbg_fc1_bwd_x_mfma(torch::Tensor dz3, torch::Tensor w3,
                                 int64_t B) {
  check_flat(dz3, "dz3"); check_flat(w3, "w3");
  auto da2 = torch::empty({B * 9216}, dz3.options());
  launch_fc1_bwd_x_mfma(dz3.data_ptr<float>(), w3.data_ptr<float>(), (int)B,
                        da2.data_ptr<float>(), cur_stream());
  return da2;
}

torch::Tensor dbg_mfma_bf16_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16 &&
              A.is_contiguous() && A.numel() == 16 * 32);
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == torch::kBFloat16 &&
              B.is_contiguous() && B.numel() == 32 * 16);
  auto D = torch::empty({16, 16},
                        A.options().dtype(torch::kFloat32));
  launch_mfma_bf16_probe(A.data_ptr(), B.data_ptr(), D.data_ptr<float>(),
                         cur_stream());
  return D;
}

// MEGA round: every sampled client of the round in ONE launch set per
// batch-step (fused_cnn_mega.hip).  Caller provides the per-client
// metadata both on device (kernel use) and host (max_batches), the
// K*P parameter/gradient stacks and a float workspace sliced here.
void cnn_round_mega(torch::Tensor shard_x, torch::Tensor shard_y,
                    torch::Tensor orders_dev, torch::Tensor row_bases_dev,
                    torch::Tensor order_offs_dev, torch::Tensor counts_dev,
                    torch::Tensor counts_host, torch::Tensor weights_dev,
                    torch::Tensor seeds_dev, int64_t bs, int64_t C,
                    torch::Tensor server_params, torch::Tensor params_stack,
                    torch::Tensor grads_stack, torch::Tensor round_accum,
                    torch::Tensor work_f, torch::Tensor work_i,
                    torch::Tensor work_b, torch::Tensor work_d,
                    torch::Tensor lr_t, double max_norm, double p1,
                    double p2, torch::Tensor stats_out,
                    torch::Tensor loss_out, bool use_bf16) {
  check_flat(server_params, "server_params");
  check_flat(params_stack, "params_stack");
  check_flat(grads_stack, "grads_stack");
  check_flat(round_accum, "round_accum");
  check_flat(work_f, "work_f");
  TORCH_CHECK(orders_dev.is_cuda() && row_bases_dev.is_cuda() &&
              order_offs_dev.is_cuda() && counts_dev.is_cuda() &&
              weights_dev.is_cuda() && seeds_dev.is_cuda(),
              "mega metadata must be device tensors");
  TORCH_CHECK(!counts_host.is_cuda() &&
              counts_host.scalar_type() == torch::kInt64);
  int K = (int)counts_host.numel();
  TORCH_CHECK(row_bases_dev.numel() == K && order_offs_dev.numel() == K &&
              counts_dev.numel() == K && weights_dev.numel() == K &&
              seeds_dev.numel() == K, "metadata lengths must equal K");
  TORCH_CHECK(bs >= 1 && bs <= 32);
  long long P = server_params.numel();
  TORCH_CHECK(params_stack.numel() >= (long long)K * P &&
              grads_stack.numel() >= (long long)K * P,
              "parameter stacks too small");
  TORCH_CHECK(stats_out.numel() >= 2 * K && loss_out.numel() >= K);
  {
    long long n_rows = shard_y.numel();
    long long n_orders = orders_dev.numel();
    auto rb = row_bases_dev.cpu();
    auto oo = order_offs_dev.cpu();
    auto rba = rb.accessor<int64_t, 1>();
    auto ooa = oo.accessor<int64_t, 1>();
    auto ct = counts_host.accessor<int64_t, 1>();
    for (int k = 0; k < K; ++k) {
      TORCH_CHECK(ct[k] > 0 && rba[k] >= 0 && ooa[k] >= 0 &&
                  rba[k] + ct[k] <= n_rows &&
                  ooa[k] + ct[k] <= n_orders,
                  "client ", k, ": metadata out of bounds");
    }
  }
  int G = K * (int)bs;
  float* f = work_f.data_ptr<float>();
  auto take = [&](long long n) { float* p = f; f += n; return p; };
  float* xb = take((long long)G * 784);
  float* a1 = take((long long)G * 21632);
  float* r2 = take((long long)G * 36864);
  float* a2 = take((long long)G * 9216);
  float* z3 = take((long long)G * 128);
  float* a3 = take((long long)G * 128);
  float* dlg = take((long long)G * C);
  float* dz3 = take((long long)G * 128);
  float* da2 = take((long long)G * 9216);
  float* dz2 = take((long long)G * 36864);
  float* dz1 = take((long long)G * 21632);
  float* w2t = take((long long)K * 18432);
  float* w2rot = take((long long)K * 18432);
  float* slab = take((long long)G * 18432);
  TORCH_CHECK(f - work_f.data_ptr<float>() <= work_f.numel(),
              "mega float workspace too small");
  TORCH_CHECK(work_i.numel() >= G, "mega int workspace too small");
  unsigned char* u = work_b.data_ptr<unsigned char>();
  unsigned char* pidx = u;
  unsigned char* m2 = u + (long long)G * 9216;
  unsigned char* m3 = u + (long long)G * 9216 * 2;
  TORCH_CHECK(work_b.numel() >= (long long)G * (9216 * 2 + 128),
              "mega byte workspace too small");
  TORCH_CHECK(work_d.numel() >= 2 * K, "mega double workspace too small");
  launch_cnn_round_mega(
      shard_x.data_ptr<float>(),
      reinterpret_cast<const long long*>(shard_y.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(orders_dev.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(row_bases_dev.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(order_offs_dev.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(counts_dev.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(counts_host.data_ptr<int64_t>()),
      weights_dev.data_ptr<float>(),
      reinterpret_cast<const long long*>(seeds_dev.data_ptr<int64_t>()),
      K, (int)bs, (int)C, server_params.data_ptr<float>(),
      params_stack.data_ptr<float>(), grads_stack.data_ptr<float>(),
      round_accum.data_ptr<float>(),
      xb, a1, r2, a2, z3, a3, dlg, dz3, da2, dz2, dz1, w2t, w2rot, slab,
      work_i.data_ptr<int>(), pidx, m2, m3, work_d.data_ptr<double>(),
      lr_t.data_ptr<float>(), (float)max_norm, (float)p1, (float)p2,
      stats_out.data_ptr<float>(), loss_out.data_ptr<float>(),
      cur_stream(), use_bf16 ? 1 : 0);
}

// per-client clip + sufficient stats + SGD over a K-stacked flat arena
// (graph-capture safe; stats_out[2k] accumulates clipped Σg / Σg²)
void mega_clip_sgd(torch::Tensor params_stack, torch::Tensor grads_stack,
                   int64_t K, torch::Tensor acc2k, double max_norm,
                   torch::Tensor lr_t, torch::Tensor stats_out) {
  check_flat(params_stack, "params_stack");
  check_flat(grads_stack, "grads_stack");
  check_flat(lr_t, "lr_t"); check_flat(stats_out, "stats_out");
  TORCH_CHECK(K > 0 && params_stack.numel() % K == 0 &&
              params_stack.numel() == grads_stack.numel());
  TORCH_CHECK(acc2k.is_cuda() && acc2k.scalar_type() == torch::kFloat64 &&
              acc2k.numel() >= 2 * K);
  TORCH_CHECK(stats_out.numel() >= 2 * K);
  long long P = params_stack.numel() / K;
  launch_mega_clip_sgd(params_stack.data_ptr<float>(),
                       grads_stack.data_ptr<float>(), P, (int)K,
                       acc2k.data_ptr<double>(), (float)max_norm,
                       lr_t.data_ptr<float>(), stats_out.data_ptr<float>(),
                       cur_stream());
}

void mega_pseudo_accum(torch::Tensor grads_stack, torch::Tensor server,
                       torch::Tensor params_stack, torch::Tensor weights_dev,
                       torch::Tensor round_accum) {
  check_flat(grads_stack, "grads_stack"); check_flat(server, "server");
  check_flat(params_stack, "params_stack");
  check_flat(weights_dev, "weights_dev");
  check_flat(round_accum, "round_accum");
  long long P = server.numel();
  TORCH_CHECK(P > 0 && params_stack.numel() % P == 0);
  int K = (int)(params_stack.numel() / P);
  TORCH_CHECK(weights_dev.numel() >= K && round_accum.numel() == P &&
              grads_stack.numel() == params_stack.numel());
  launch_mega_pseudo_accum(grads_stack.data_ptr<float>(),
                           server.data_ptr<float>(),
                           params_stack.data_ptr<float>(),
                           weights_dev.data_ptr<float>(),
                           round_accum.data_ptr<float>(), P, K,
                           cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("cnn_round_mega", &cnn_round_mega);
  m.def("mega_clip_sgd", &mega_clip_sgd);
  m.def("mega_pseudo_accum", &mega_pseudo_accum);
  m.def("dbg_mfma_bf16_probe", &dbg_mfma_bf16_probe);
  m.def("dbg_conv2_fwd_mfma_bf16", &dbg_conv2_fwd_mfma_bf16);
  m.def("dbg_conv2_bwd_x_mfma_bf16", &dbg_conv2_bwd_x_mfma_bf16);
  m.def("dbg_conv2_bwd_w_mfma_bf16", &dbg_conv2_bwd_w_mfma_bf16);
  m.def("dbg_fc1_fwd_mfma_bf16", &dbg_fc1_fwd_mfma_bf16);
  m.def("dbg_conv2_fwd_mfma", &dbg_conv2_fwd_mfma);
  m.def("dbg_conv2_bwd_x_mfma", &dbg_conv2_bwd_x_mfma);
  m.def("dbg_conv2_bwd_w_mfma", &dbg_conv2_bwd_w_mfma);
  m.def("dbg_fc1_fwd_mfma", &dbg_fc1_fwd_mfma);
  m.def("dbg_fc1_bwd_w_mfma", &dbg_fc1_bwd_w_mfma);
  m.def("dbg_fc1_bwd_x_mfma", &dbg_fc1_bwd_x_mfma);
  m.doc() = "msrflute_amd gfx950 flat-arena kernels";
  m.def("pseudo_grad", &pseudo_grad);
  m.def("axpy", &axpy);
  m.def("scale", &scale);
  m.def("sum_sumsq", &sum_sumsq);
  m.def("clip_by_norm", &clip_by_norm);
  m.def("add_gaussian_noise", &add_gaussian_noise);
  m.def("sgd_step", &sgd_step);
  m.def("sgd_step_devlr", &sgd_step_devlr);
  m.def("clip_stats_accumulate", &clip_stats_accumulate);
  m.def("adam_step", &adam_step);
  m.def("adamax_step", &adamax_step);
  m.def("segmented_sqnorm", &segmented_sqnorm);
  m.def("quant_bin_mask", &quant_bin_mask);
  m.def("gru_gates", &gru_gates);
  m.def("cnn_epoch", &cnn_epoch);
  m.def("cnn_round", &cnn_round);
  m.def("lstm_seq_fwd", &lstm_seq_fwd);
  m.def("lstm_seq_fwd_b", &lstm_seq_fwd_b);
  m.def("lstm_seq_bwd_b", &lstm_seq_bwd_b);
  m.def("gru_seq_fwd", &gru_seq_fwd);
  m.def("gru_seq_bwd", &gru_seq_bwd);
  m.def("lstm_seq_bwd", &lstm_seq_bwd);
}
