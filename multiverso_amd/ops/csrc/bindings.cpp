// Torch binding layer for the gfx950 HIP kernels (kernels.hip).
//
// This translation unit is compiled by torch.utils.cpp_extension (it needs
// the ATen headers); the kernels themselves are hipcc-compiled pure HIP in
// kernels.hip and linked in as an object, so no hipify/CUDA-compat layer
// ever touches the device code.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstdint>

extern "C" {
void mv_launch_copy(float*, const float*, int64_t, hipStream_t);
void mv_launch_add(float*, const float*, int64_t, hipStream_t);
void mv_launch_sgd(float*, const float*, int64_t, hipStream_t);
void mv_launch_momentum(float*, float*, const float*, float, int64_t, hipStream_t);
void mv_launch_adagrad(float*, float*, const float*, float, float, float,
                       int64_t, hipStream_t);
void mv_launch_sgd_copy(float*, const float*, float*, float, int64_t,
                        hipStream_t);
void mv_launch_momentum_copy(float*, float*, const float*, float*, float,
                             int64_t, hipStream_t);
void mv_launch_adagrad_copy(float*, float*, const float*, float*, float,
                            float, float, int64_t, hipStream_t);
void mv_launch_dcasgd_copy(float*, float*, const float*, float*, float,
                           float, int64_t, hipStream_t);
void mv_launch_dcasgda_copy(float*, float*, float*, const float*, float*,
                            float, float, float, float, int64_t,
                            hipStream_t);
void mv_launch_dcasgd(float*, float*, const float*, float, float, int64_t,
                      hipStream_t);
void mv_launch_dcasgda(float*, float*, float*, const float*, float, float,
                       float, float, int64_t, hipStream_t);
void mv_launch_row_gather(float*, const float*, const int64_t*, int64_t,
                          int64_t, hipStream_t);
void mv_launch_row_scatter_add(float*, const float*, const int64_t*, float,
                               int64_t, int64_t, int, hipStream_t);
void mv_launch_w2v(float*, float*, float*, float*, const int64_t*, const int*,
                   const int64_t*, const float*, const int*, float, int64_t,
                   int64_t, int, int, hipStream_t);
void mv_launch_w2v_ns(float*, float*, float*, float*, const int64_t*,
                      const int*, const int64_t*, const int64_t*, int64_t,
                      int, uint64_t, float, int64_t, int64_t, int, int,
                      hipStream_t);
void mv_launch_lr_sigmoid_fwd(const float*, const int64_t*, const float*,
                              const int*, const float*, const float*,
                              float*, float*, int64_t, hipStream_t);
void mv_launch_lr_sigmoid_scatter(float*, const int64_t*, const float*,
                                  const int*, const float*, float, int,
                                  float, int64_t, hipStream_t);
void mv_launch_row_scatter_adagrad(float*, float*, const float*,
                                   const int64_t*, float, float, float,
                                   int64_t, int64_t, int, hipStream_t);
void mv_launch_lr_softmax_fwd(const float*, const int64_t*, const float*,
                              const int*, const float*, const float*,
                              float*, float*, int64_t, int64_t, hipStream_t);
void mv_launch_lr_softmax_scatter(float*, const int64_t*, const float*,
                                  const int*, const float*, float, int,
                                  float, int64_t, int64_t, hipStream_t);
void mv_launch_lr_dense_post(float*, const float*, const float*, float*,
                             float, int64_t, int64_t, hipStream_t);
int mv_launch_lr_dense_fwd(const float*, const float*, const float*,
                           const float*, float*, float*, float, int64_t,
                           int64_t, int64_t, int, hipStream_t);
void mv_launch_lr_ftrl_fwd(const float*, const int64_t*, const float*,
                           const int*, const float*, const float*, float*,
                           float*, float, float, float, float, int64_t,
                           int64_t, hipStream_t);
void mv_launch_lr_ftrl_scatter(float*, const int64_t*, const float*,
                               const int*, const float*, float, float,
                               float, float, int64_t, int64_t, hipStream_t);
void mv_launch_add_f64(double*, const double*, int64_t, hipStream_t);
void mv_launch_sgd_f64(double*, const double*, int64_t, hipStream_t);
void mv_launch_momentum_f64(double*, double*, const double*, double, int64_t,
                            hipStream_t);
void mv_launch_adagrad_f64(double*, double*, const double*, double, double,
                           double, int64_t, hipStream_t);
void mv_launch_sgd_copy_f64(double*, const double*, double*, double, int64_t,
                            hipStream_t);
void mv_launch_add_i32(int32_t*, const int32_t*, int64_t, hipStream_t);
void mv_launch_add_i64(int64_t*, const int64_t*, int64_t, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

void check_dev(const torch::Tensor& t, const char* name,
               torch::ScalarType st) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == st, name, " dtype mismatch");
}

void nt_copy(torch::Tensor dst, torch::Tensor src) {
  check_f32(dst, "dst"); check_f32(src, "src");
  TORCH_CHECK(dst.numel() == src.numel(), "size mismatch");
  mv_launch_copy(dst.data_ptr<float>(), src.data_ptr<float>(), dst.numel(),
                 cur_stream());
}

void add_inplace(torch::Tensor data, torch::Tensor delta) {
  // dtype-dispatched K1 (reference instantiates int/float/double tables,
  // array_table.cpp:153-154; int updater is add-only, updater.cpp:40-43)
  auto st = data.scalar_type();
  check_dev(data, "data", st); check_dev(delta, "delta", st);
  TORCH_CHECK(data.numel() == delta.numel(), "size mismatch");
  int64_t n = data.numel();
  if (st == torch::kFloat32)
    mv_launch_add(data.data_ptr<float>(), delta.data_ptr<float>(), n,
                  cur_stream());
  else if (st == torch::kFloat64)
    mv_launch_add_f64(data.data_ptr<double>(), delta.data_ptr<double>(), n,
                      cur_stream());
  else if (st == torch::kInt32)
    mv_launch_add_i32(data.data_ptr<int32_t>(), delta.data_ptr<int32_t>(), n,
                      cur_stream());
  else if (st == torch::kInt64)
    mv_launch_add_i64(data.data_ptr<int64_t>(), delta.data_ptr<int64_t>(), n,
                      cur_stream());
  else
    TORCH_CHECK(false, "add_inplace: unsupported dtype ", data.dtype());
}

void sgd_update(torch::Tensor data, torch::Tensor delta) {
  auto st = data.scalar_type();
  check_dev(data, "data", st); check_dev(delta, "delta", st);
  TORCH_CHECK(data.numel() == delta.numel(), "size mismatch");
  if (st == torch::kFloat32)
    mv_launch_sgd(data.data_ptr<float>(), delta.data_ptr<float>(),
                  data.numel(), cur_stream());
  else if (st == torch::kFloat64)
    mv_launch_sgd_f64(data.data_ptr<double>(), delta.data_ptr<double>(),
                      data.numel(), cur_stream());
  else
    TORCH_CHECK(false, "sgd_update: unsupported dtype ", data.dtype());
}

void momentum_update(torch::Tensor data, torch::Tensor m, torch::Tensor delta,
                     double mu) {
  auto st = data.scalar_type();
  check_dev(data, "data", st); check_dev(m, "m", st);
  check_dev(delta, "delta", st);
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == m.numel(),
              "size mismatch");
  if (st == torch::kFloat32)
    mv_launch_momentum(data.data_ptr<float>(), m.data_ptr<float>(),
                       delta.data_ptr<float>(), (float)mu, data.numel(),
                       cur_stream());
  else if (st == torch::kFloat64)
    mv_launch_momentum_f64(data.data_ptr<double>(), m.data_ptr<double>(),
                           delta.data_ptr<double>(), mu, data.numel(),
                           cur_stream());
  else
    TORCH_CHECK(false, "momentum_update: unsupported dtype ", data.dtype());
}

void adagrad_update(torch::Tensor data, torch::Tensor gsq, torch::Tensor delta,
                    double lr, double rho, double eps) {
  auto st = data.scalar_type();
  check_dev(data, "data", st); check_dev(gsq, "gsq", st);
  check_dev(delta, "delta", st);
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == gsq.numel(),
              "size mismatch");
  if (st == torch::kFloat32)
    mv_launch_adagrad(data.data_ptr<float>(), gsq.data_ptr<float>(),
                      delta.data_ptr<float>(), (float)lr, (float)rho,
                      (float)eps, data.numel(), cur_stream());
  else if (st == torch::kFloat64)
    mv_launch_adagrad_f64(data.data_ptr<double>(), gsq.data_ptr<double>(),
                          delta.data_ptr<double>(), lr, rho, eps,
                          data.numel(), cur_stream());
  else
    TORCH_CHECK(false, "adagrad_update: unsupported dtype ", data.dtype());
}

void sgd_copy_update(torch::Tensor data, torch::Tensor delta,
                     torch::Tensor out, double sign) {
  auto st = data.scalar_type();
  check_dev(data, "data", st); check_dev(delta, "delta", st);
  check_dev(out, "out", st);
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == out.numel(),
              "size mismatch");
  if (st == torch::kFloat32)
    mv_launch_sgd_copy(data.data_ptr<float>(), delta.data_ptr<float>(),
                       out.data_ptr<float>(), (float)sign, data.numel(),
                       cur_stream());
  else if (st == torch::kFloat64)
    mv_launch_sgd_copy_f64(data.data_ptr<double>(), delta.data_ptr<double>(),
                           out.data_ptr<double>(), sign, data.numel(),
                           cur_stream());
  else
    TORCH_CHECK(false, "sgd_copy_update: unsupported dtype ", data.dtype());
}

void momentum_copy_update(torch::Tensor data, torch::Tensor m,
                          torch::Tensor delta, torch::Tensor out, double mu) {
  check_f32(data, "data"); check_f32(m, "m"); check_f32(delta, "delta");
  check_f32(out, "out");
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == m.numel() &&
              data.numel() == out.numel(), "size mismatch");
  mv_launch_momentum_copy(data.data_ptr<float>(), m.data_ptr<float>(),
                          delta.data_ptr<float>(), out.data_ptr<float>(),
                          (float)mu, data.numel(), cur_stream());
}

void adagrad_copy_update(torch::Tensor data, torch::Tensor gsq,
                         torch::Tensor delta, torch::Tensor out,
                         double lr, double rho, double eps) {
  check_f32(data, "data"); check_f32(gsq, "gsq"); check_f32(delta, "delta");
  check_f32(out, "out");
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == gsq.numel() &&
              data.numel() == out.numel(), "size mismatch");
  mv_launch_adagrad_copy(data.data_ptr<float>(), gsq.data_ptr<float>(),
                         delta.data_ptr<float>(), out.data_ptr<float>(),
                         (float)lr, (float)rho, (float)eps, data.numel(),
                         cur_stream());
}

void dcasgd_copy_update(torch::Tensor data, torch::Tensor bak,
                        torch::Tensor delta, torch::Tensor out,
                        double lr, double lambda) {
  check_f32(data, "data"); check_f32(bak, "bak"); check_f32(delta, "delta");
  check_f32(out, "out");
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == bak.numel() &&
              data.numel() == out.numel(), "size mismatch");
  mv_launch_dcasgd_copy(data.data_ptr<float>(), bak.data_ptr<float>(),
                        delta.data_ptr<float>(), out.data_ptr<float>(),
                        (float)lr, (float)lambda, data.numel(), cur_stream());
}

void dcasgda_copy_update(torch::Tensor data, torch::Tensor bak,
                         torch::Tensor msq, torch::Tensor delta,
                         torch::Tensor out, double lr, double lambda,
                         double rho, double eps) {
  check_f32(data, "data"); check_f32(bak, "bak"); check_f32(msq, "msq");
  check_f32(delta, "delta"); check_f32(out, "out");
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == bak.numel() &&
              data.numel() == msq.numel() && data.numel() == out.numel(),
              "size mismatch");
  mv_launch_dcasgda_copy(data.data_ptr<float>(), bak.data_ptr<float>(),
                         msq.data_ptr<float>(), delta.data_ptr<float>(),
                         out.data_ptr<float>(), (float)lr, (float)lambda,
                         (float)rho, (float)eps, data.numel(), cur_stream());
}

void dcasgd_update(torch::Tensor data, torch::Tensor bak, torch::Tensor delta,
                   double lr, double lambda) {
  check_f32(data, "data"); check_f32(bak, "bak"); check_f32(delta, "delta");
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == bak.numel(),
              "size mismatch");
  mv_launch_dcasgd(data.data_ptr<float>(), bak.data_ptr<float>(),
                   delta.data_ptr<float>(), (float)lr, (float)lambda,
                   data.numel(), cur_stream());
}

void dcasgda_update(torch::Tensor data, torch::Tensor bak, torch::Tensor msq,
                    torch::Tensor delta, double lr, double lambda, double rho,
                    double eps) {
  check_f32(data, "data"); check_f32(bak, "bak"); check_f32(msq, "msq");
  check_f32(delta, "delta");
  TORCH_CHECK(data.numel() == delta.numel() && data.numel() == bak.numel() &&
              data.numel() == msq.numel(), "size mismatch");
  mv_launch_dcasgda(data.data_ptr<float>(), bak.data_ptr<float>(),
                    msq.data_ptr<float>(), delta.data_ptr<float>(), (float)lr,
                    (float)lambda, (float)rho, (float)eps, data.numel(),
                    cur_stream());
}

torch::Tensor row_gather(torch::Tensor shard, torch::Tensor rows) {
  check_f32(shard, "shard");
  TORCH_CHECK(shard.dim() == 2, "shard must be 2-D");
  TORCH_CHECK(rows.scalar_type() == torch::kInt64 && rows.is_cuda() &&
              rows.is_contiguous(), "rows must be contiguous int64 on GPU");
  auto out = torch::empty({rows.numel(), shard.size(1)}, shard.options());
  mv_launch_row_gather(out.data_ptr<float>(), shard.data_ptr<float>(),
                       rows.data_ptr<int64_t>(), rows.numel(), shard.size(1),
                       cur_stream());
  return out;
}

void row_gather_out(torch::Tensor out, torch::Tensor shard, torch::Tensor rows) {
  check_f32(shard, "shard"); check_f32(out, "out");
  TORCH_CHECK(shard.dim() == 2, "shard must be 2-D");
  TORCH_CHECK(rows.scalar_type() == torch::kInt64 && rows.is_cuda() &&
              rows.is_contiguous(), "rows must be contiguous int64 on GPU");
  TORCH_CHECK(out.numel() == rows.numel() * shard.size(1), "out size mismatch");
  mv_launch_row_gather(out.data_ptr<float>(), shard.data_ptr<float>(),
                       rows.data_ptr<int64_t>(), rows.numel(), shard.size(1),
                       cur_stream());
}

void row_scatter_add(torch::Tensor shard, torch::Tensor rows,
                     torch::Tensor vals, double sign,
                     bool assume_unique = false) {
  check_f32(shard, "shard"); check_f32(vals, "vals");
  TORCH_CHECK(shard.dim() == 2, "shard must be 2-D");
  TORCH_CHECK(rows.scalar_type() == torch::kInt64 && rows.is_cuda() &&
              rows.is_contiguous(), "rows must be contiguous int64 on GPU");
  TORCH_CHECK(vals.numel() == rows.numel() * shard.size(1), "vals size mismatch");
  mv_launch_row_scatter_add(shard.data_ptr<float>(), vals.data_ptr<float>(),
                            rows.data_ptr<int64_t>(), (float)sign,
                            rows.numel(), shard.size(1),
                            assume_unique ? 1 : 0, cur_stream());
}

void row_scatter_adagrad(torch::Tensor shard, torch::Tensor gsq,
                         torch::Tensor rows, torch::Tensor vals,
                         double lr, double rho, double eps,
                         bool assume_unique = false) {
  check_f32(shard, "shard"); check_f32(gsq, "gsq"); check_f32(vals, "vals");
  TORCH_CHECK(shard.dim() == 2, "shard must be 2-D");
  TORCH_CHECK(gsq.sizes() == shard.sizes(), "gsq shape mismatch");
  TORCH_CHECK(rows.scalar_type() == torch::kInt64 && rows.is_cuda() &&
              rows.is_contiguous(), "rows must be contiguous int64 on GPU");
  TORCH_CHECK(vals.numel() == rows.numel() * shard.size(1), "vals mismatch");
  mv_launch_row_scatter_adagrad(
      shard.data_ptr<float>(), gsq.data_ptr<float>(), vals.data_ptr<float>(),
      rows.data_ptr<int64_t>(), (float)lr, (float)rho, (float)eps,
      rows.numel(), shard.size(1), assume_unique ? 1 : 0, cur_stream());
}

void w2v_train(torch::Tensor in_emb, torch::Tensor out_emb,
               torch::Tensor in_gsq, torch::Tensor out_gsq,
               torch::Tensor in_idx, torch::Tensor in_off,
               torch::Tensor out_idx, torch::Tensor out_label,
               torch::Tensor out_off, double lr, bool use_adagrad,
               bool use_atomic) {
  check_f32(in_emb, "in_emb"); check_f32(out_emb, "out_emb");
  check_f32(out_label, "out_label");
  TORCH_CHECK(in_emb.dim() == 2 && out_emb.dim() == 2, "emb must be 2-D");
  TORCH_CHECK(in_emb.size(1) == out_emb.size(1), "dim mismatch");
  TORCH_CHECK(in_emb.size(1) <= 2048, "w2v kernel supports dim <= 2048");
  TORCH_CHECK(in_idx.scalar_type() == torch::kInt64 &&
              out_idx.scalar_type() == torch::kInt64, "idx must be int64");
  TORCH_CHECK(in_off.scalar_type() == torch::kInt32 &&
              out_off.scalar_type() == torch::kInt32, "offsets must be int32");
  int64_t G = in_off.numel() - 1;
  TORCH_CHECK(out_off.numel() - 1 == G, "group count mismatch");
  float *igq = nullptr, *ogq = nullptr;
  if (use_adagrad) {
    check_f32(in_gsq, "in_gsq"); check_f32(out_gsq, "out_gsq");
    TORCH_CHECK(in_gsq.sizes() == in_emb.sizes() &&
                out_gsq.sizes() == out_emb.sizes(), "gsq shape mismatch");
    igq = in_gsq.data_ptr<float>();
    ogq = out_gsq.data_ptr<float>();
  }
  mv_launch_w2v(in_emb.data_ptr<float>(), out_emb.data_ptr<float>(), igq, ogq,
                in_idx.data_ptr<int64_t>(), in_off.data_ptr<int>(),
                out_idx.data_ptr<int64_t>(), out_label.data_ptr<float>(),
                out_off.data_ptr<int>(), (float)lr, G, in_emb.size(1),
                use_adagrad ? 1 : 0, use_atomic ? 1 : 0, cur_stream());
}

void w2v_train_ns(torch::Tensor in_emb, torch::Tensor out_emb,
                  torch::Tensor in_gsq, torch::Tensor out_gsq,
                  torch::Tensor in_idx,
                  c10::optional<torch::Tensor> in_off_opt,
                  torch::Tensor centers, torch::Tensor pool, int64_t neg,
                  int64_t seed, double lr, bool use_adagrad,
                  bool use_atomic) {
  check_f32(in_emb, "in_emb"); check_f32(out_emb, "out_emb");
  TORCH_CHECK(in_emb.dim() == 2 && out_emb.dim() == 2, "emb must be 2-D");
  TORCH_CHECK(in_emb.size(1) == out_emb.size(1), "dim mismatch");
  TORCH_CHECK(in_emb.size(1) <= 2048, "w2v kernel supports dim <= 2048");
  TORCH_CHECK(in_idx.scalar_type() == torch::kInt64 &&
              centers.scalar_type() == torch::kInt64 &&
              pool.scalar_type() == torch::kInt64, "ids must be int64");
  const int* in_off_ptr = nullptr;
  if (in_off_opt.has_value()) {
    TORCH_CHECK(in_off_opt->scalar_type() == torch::kInt32,
                "offsets must be int32");
    TORCH_CHECK(in_off_opt->numel() - 1 == centers.numel(),
                "centers/group count mismatch");
    in_off_ptr = in_off_opt->data_ptr<int>();
  } else {
    // skip-gram: one input per group (in_idx[g] is group g's context)
    TORCH_CHECK(in_idx.numel() == centers.numel(),
                "skip-gram in_idx/centers mismatch");
  }
  TORCH_CHECK(pool.numel() > 0, "empty negative pool");
  int64_t G = centers.numel();
  float *igq = nullptr, *ogq = nullptr;
  if (use_adagrad) {
    check_f32(in_gsq, "in_gsq"); check_f32(out_gsq, "out_gsq");
    TORCH_CHECK(in_gsq.sizes() == in_emb.sizes() &&
                out_gsq.sizes() == out_emb.sizes(), "gsq shape mismatch");
    igq = in_gsq.data_ptr<float>();
    ogq = out_gsq.data_ptr<float>();
  }
  mv_launch_w2v_ns(in_emb.data_ptr<float>(), out_emb.data_ptr<float>(), igq,
                   ogq, in_idx.data_ptr<int64_t>(), in_off_ptr,
                   centers.data_ptr<int64_t>(), pool.data_ptr<int64_t>(),
                   pool.numel(), (int)neg, (uint64_t)seed, (float)lr, G,
                   in_emb.size(1), use_adagrad ? 1 : 0, use_atomic ? 1 : 0,
                   cur_stream());
}

void lr_sigmoid_forward(torch::Tensor w, torch::Tensor keys,
                        torch::Tensor vals, torch::Tensor ptr,
                        torch::Tensor labels,
                        c10::optional<torch::Tensor> wts,
                        torch::Tensor err, torch::Tensor loss) {
  check_f32(w, "w"); check_f32(vals, "vals"); check_f32(labels, "labels");
  check_f32(err, "err"); check_f32(loss, "loss");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
  int64_t B = ptr.numel() - 1;
  TORCH_CHECK(labels.numel() == B && err.numel() == B && loss.numel() == B,
              "batch size mismatch");
  TORCH_CHECK(keys.numel() == vals.numel(), "keys/vals mismatch");
  const float* wp = nullptr;
  if (wts.has_value()) {
    check_f32(*wts, "wts");
    TORCH_CHECK(wts->numel() == B, "weights size mismatch");
    wp = wts->data_ptr<float>();
  }
  mv_launch_lr_sigmoid_fwd(w.data_ptr<float>(), keys.data_ptr<int64_t>(),
                           vals.data_ptr<float>(), ptr.data_ptr<int>(),
                           labels.data_ptr<float>(), wp,
                           err.data_ptr<float>(), loss.data_ptr<float>(), B,
                           cur_stream());
}

void lr_sigmoid_scatter(torch::Tensor w, torch::Tensor keys,
                        torch::Tensor vals, torch::Tensor ptr,
                        torch::Tensor err, double lr, int64_t reg_type,
                        double reg_coef) {
  check_f32(w, "w"); check_f32(vals, "vals"); check_f32(err, "err");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
  int64_t B = ptr.numel() - 1;
  TORCH_CHECK(err.numel() == B, "err size mismatch");
  mv_launch_lr_sigmoid_scatter(w.data_ptr<float>(), keys.data_ptr<int64_t>(),
                               vals.data_ptr<float>(), ptr.data_ptr<int>(),
                               err.data_ptr<float>(), (float)lr,
                               (int)reg_type, (float)reg_coef, B,
                               cur_stream());
}

void lr_softmax_forward(torch::Tensor w, torch::Tensor keys,
                        torch::Tensor vals, torch::Tensor ptr,
                        torch::Tensor labels,
                        c10::optional<torch::Tensor> wts,
                        torch::Tensor err, torch::Tensor loss, int64_t K) {
  check_f32(w, "w"); check_f32(vals, "vals"); check_f32(labels, "labels");
  check_f32(err, "err"); check_f32(loss, "loss");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
  TORCH_CHECK(K >= 2 && K <= 64,
              "fused softmax supports 2..64 classes (got ", K, ")");
  int64_t B = ptr.numel() - 1;
  TORCH_CHECK(labels.numel() == B && loss.numel() == B, "batch mismatch");
  TORCH_CHECK(err.numel() == B * K, "err must be [B, K]");
  TORCH_CHECK(keys.numel() == vals.numel(), "keys/vals mismatch");
  const float* wp = nullptr;
  if (wts.has_value()) {
    check_f32(*wts, "wts");
    TORCH_CHECK(wts->numel() == B, "weights size mismatch");
    wp = wts->data_ptr<float>();
  }
  mv_launch_lr_softmax_fwd(w.data_ptr<float>(), keys.data_ptr<int64_t>(),
                           vals.data_ptr<float>(), ptr.data_ptr<int>(),
                           labels.data_ptr<float>(), wp,
                           err.data_ptr<float>(), loss.data_ptr<float>(),
                           B, K, cur_stream());
}

void lr_softmax_scatter(torch::Tensor w, torch::Tensor keys,
                        torch::Tensor vals, torch::Tensor ptr,
                        torch::Tensor err, double lr, int64_t reg_type,
                        double reg_coef, int64_t K) {
  check_f32(w, "w"); check_f32(vals, "vals"); check_f32(err, "err");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
  int64_t B = ptr.numel() - 1;
  TORCH_CHECK(err.numel() == B * K, "err must be [B, K]");
  mv_launch_lr_softmax_scatter(w.data_ptr<float>(), keys.data_ptr<int64_t>(),
                               vals.data_ptr<float>(), ptr.data_ptr<int>(),
                               err.data_ptr<float>(), (float)lr,
                               (int)reg_type, (float)reg_coef, B, K,
                               cur_stream());
}

void lr_ftrl_forward(torch::Tensor zn, torch::Tensor keys,
                     torch::Tensor vals, torch::Tensor ptr,
                     torch::Tensor labels, c10::optional<torch::Tensor> wts,
                     torch::Tensor err, torch::Tensor loss, double alpha,
                     double beta, double l1, double l2, int64_t K) {
  check_f32(zn, "zn"); check_f32(vals, "vals"); check_f32(labels, "labels");
  check_f32(err, "err"); check_f32(loss, "loss");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
  TORCH_CHECK(K >= 1 && K <= 32,
              "fused FTRL supports 1..32 outputs (got ", K, ")");
  int64_t B = ptr.numel() - 1;
  TORCH_CHECK(labels.numel() == B && loss.numel() == B, "batch mismatch");
  TORCH_CHECK(err.numel() == B * K, "err must be [B, K]");
  const float* wp = nullptr;
  if (wts.has_value()) {
    check_f32(*wts, "wts");
    wp = wts->data_ptr<float>();
  }
  mv_launch_lr_ftrl_fwd(zn.data_ptr<float>(), keys.data_ptr<int64_t>(),
                        vals.data_ptr<float>(), ptr.data_ptr<int>(),
                        labels.data_ptr<float>(), wp, err.data_ptr<float>(),
                        loss.data_ptr<float>(), (float)(1.0 / alpha),
                        (float)beta, (float)l1, (float)l2, B, K,
                        cur_stream());
}

void lr_dense_post(torch::Tensor logits, torch::Tensor labels,
                   c10::optional<torch::Tensor> wts, torch::Tensor loss_acc,
                   double inv_b) {
  check_f32(logits, "logits"); check_f32(labels, "labels");
  check_f32(loss_acc, "loss_acc");
  int64_t B = labels.numel();
  TORCH_CHECK(B > 0 && logits.numel() % B == 0,
              "logits/labels batch mismatch");
  int64_t K = logits.numel() / B;
  TORCH_CHECK(K >= 1 && K <= 64,
              "fused dense post supports 1..64 classes (got ", K, ")");
  TORCH_CHECK(loss_acc.numel() == 1, "loss_acc must be a scalar");
  const float* wp = nullptr;
  if (wts.has_value()) {
    check_f32(*wts, "wts");
    TORCH_CHECK(wts->numel() == B, "weights size mismatch");
    wp = wts->data_ptr<float>();
  }
  mv_launch_lr_dense_post(logits.data_ptr<float>(),
                          labels.data_ptr<float>(), wp,
                          loss_acc.data_ptr<float>(), (float)inv_b, B, K,
                          cur_stream());
}

bool lr_dense_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor labels,
                  c10::optional<torch::Tensor> wts, torch::Tensor diff,
                  torch::Tensor loss_acc, double inv_b, bool nt) {
  check_f32(x, "x"); check_f32(w, "w"); check_f32(labels, "labels");
  check_f32(diff, "diff"); check_f32(loss_acc, "loss_acc");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.size(1) == w.size(0),
              "x [B,d] and w [d,K] shape mismatch");
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous(), "x/w must be "
              "contiguous");
  int64_t B = x.size(0), d = x.size(1), K = w.size(1);
  TORCH_CHECK(labels.numel() == B, "labels batch mismatch");
  TORCH_CHECK(diff.numel() == B * K, "diff must be [B, K]");
  TORCH_CHECK(loss_acc.numel() == 1, "loss_acc must be a scalar");
  const float* wp = nullptr;
  if (wts.has_value()) {
    check_f32(*wts, "wts");
    TORCH_CHECK(wts->numel() == B, "weights size mismatch");
    wp = wts->data_ptr<float>();
  }
  return mv_launch_lr_dense_fwd(
             x.data_ptr<float>(), w.data_ptr<float>(),
             labels.data_ptr<float>(), wp, diff.data_ptr<float>(),
             loss_acc.data_ptr<float>(), (float)inv_b, B, d, K,
             nt ? 1 : 0, cur_stream()) != 0;
}


void lr_ftrl_scatter(torch::Tensor zn, torch::Tensor keys,
                     torch::Tensor vals, torch::Tensor ptr,
                     torch::Tensor err, double alpha, double beta, double l1,
                     double l2, int64_t K) {
  check_f32(zn, "zn"); check_f32(vals, "vals"); check_f32(err, "err");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
  int64_t B = ptr.numel() - 1;
  TORCH_CHECK(err.numel() == B * K, "err must be [B, K]");
  mv_launch_lr_ftrl_scatter(zn.data_ptr<float>(), keys.data_ptr<int64_t>(),
                            vals.data_ptr<float>(), ptr.data_ptr<int>(),
                            err.data_ptr<float>(), (float)(1.0 / alpha),
                            (float)beta, (float)l1, (float)l2, B, K,
                            cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("add_inplace", &add_inplace, "K1: data += delta (fp32, fused)");
  m.def("nt_copy", &nt_copy, "K7: non-temporal streaming copy");
  m.def("sgd_update", &sgd_update, "K2: data -= delta");
  m.def("momentum_update", &momentum_update, "K3: fused momentum update");
  m.def("adagrad_update", &adagrad_update, "K4: fused adagrad update");
  m.def("momentum_copy_update", &momentum_copy_update,
        "fused momentum Add+Get");
  m.def("adagrad_copy_update", &adagrad_copy_update,
        "fused adagrad Add+Get");
  m.def("dcasgd_copy_update", &dcasgd_copy_update,
        "fused dcasgd Add+Get");
  m.def("dcasgda_copy_update", &dcasgda_copy_update,
        "fused dcasgda Add+Get");
  m.def("sgd_copy_update", &sgd_copy_update,
        "fused Add+Get: data (+/-)= delta; out = data (saves the Get's "
        "shard re-read at N=1)");
  m.def("dcasgd_update", &dcasgd_update,
        "DC-ASGD: delay-compensated update with per-worker backup");
  m.def("dcasgda_update", &dcasgda_update,
        "DC-ASGD-a: adaptive lambda via mean-square of gradients");
  m.def("row_gather", &row_gather, "K6: out[i] = shard[rows[i]]");
  m.def("row_gather_out", &row_gather_out, "K6 (preallocated out)");
  m.def("row_scatter_add", &row_scatter_add,
        "K5: shard[rows[i]] += sign*vals[i] (atomic; assume_unique=True "
        "uses plain stores)", py::arg("shard"), py::arg("rows"),
        py::arg("vals"), py::arg("sign"), py::arg("assume_unique") = false);
  m.def("lr_sigmoid_forward", &lr_sigmoid_forward,
        "K13 fused: per-sample CSR dot + sigmoid + error/loss");
  m.def("lr_sigmoid_scatter", &lr_sigmoid_scatter,
        "K14 fused: w[keys] -= lr*(vals*err + reg), atomic");
  m.def("lr_softmax_forward", &lr_softmax_forward,
        "K13 softmax fused: per-sample CSR K-class logits + softmax + "
        "err/loss (objective.cpp:193-230)");
  m.def("lr_softmax_scatter", &lr_softmax_scatter,
        "K14 softmax fused: w[key*K+k] -= lr*(val*err_k + reg), atomic");
  m.def("lr_dense_fwd", &lr_dense_fwd,
        "Dense-mode fused forward: X@W + softmax/sigmoid diff + loss in "
        "one kernel, W LDS-staged; returns False when d*K exceeds the "
        "LDS budget (caller uses the GEMM + lr_dense_post path)",
        py::arg("x"), py::arg("w"), py::arg("labels"), py::arg("wts"),
        py::arg("diff"), py::arg("loss_acc"), py::arg("inv_b"),
        py::arg("nt") = true);
  m.def("lr_dense_post", &lr_dense_post,
        "Dense-mode fused post-GEMM: logits -> softmax/sigmoid diff in "
        "place + atomic mean-loss accumulate (objective.cpp:193-230 "
        "dense branch)");
  m.def("lr_ftrl_forward", &lr_ftrl_forward,
        "FTRL fused forward: reconstruct w from (z|n), sigmoid, err/loss "
        "(objective.cpp:250-345)");
  m.def("lr_ftrl_scatter", &lr_ftrl_scatter,
        "FTRL fused z/n state update (updater.cpp:79-101 applied to the "
        "local chunk buffer)");
  m.def("row_scatter_adagrad", &row_scatter_adagrad,
        "K15: keyed adagrad update on owned shard rows "
        "(assume_unique=True uses plain stores)",
        py::arg("shard"), py::arg("gsq"), py::arg("rows"), py::arg("vals"),
        py::arg("lr"), py::arg("rho"), py::arg("eps"),
        py::arg("assume_unique") = false);
  m.def("w2v_train", &w2v_train,
        "K9-K11: fused word2vec block training (skip-gram/CBOW, NS/HS, "
        "optional adagrad)");
  m.def("w2v_train_ns", &w2v_train_ns,
        "K9-K11 NS fast path: negatives generated in-kernel from the "
        "per-block pool (reference LCG scheme)");
}
