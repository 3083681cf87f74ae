// Python bindings for the MI355X PS kernels (ps_kernels.hip) and GEMM
// (gemm_bf16.hip). Thin argument-checking layer: all compute stays in the
// HIP translation units; this file never includes HIP headers so it
// compiles as plain C++ against libtorch.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include <cstdint>

extern "C" {
void edl_dense_sgd(float*, const float*, int64_t, float, hipStream_t);
void edl_dense_momentum(float*, float*, const float*, int64_t, float, float,
                        bool, hipStream_t);
void edl_dense_adam(float*, float*, float*, float*, const float*, int64_t,
                    float, float, float, float, hipStream_t);
void edl_dense_adagrad(float*, float*, const float*, int64_t, float, float,
                       hipStream_t);
void edl_dense_ftrl(float*, float*, float*, const float*, int64_t, float,
                    float, float, float, hipStream_t);
void edl_sparse_sgd(float*, const float*, const int32_t*, int64_t, int64_t,
                    float, const int32_t*, hipStream_t);
void edl_sparse_momentum(float*, float*, const float*, const int32_t*,
                         int64_t, int64_t, float, float, bool,
                         const int32_t*, hipStream_t);
void edl_sparse_adam(float*, float*, float*, float*, const float*,
                     const int32_t*, int64_t, int64_t, float, float, float,
                     float, const int32_t*, hipStream_t);
void edl_sparse_adagrad(float*, float*, const float*, const int32_t*, int64_t,
                        int64_t, float, float, const int32_t*, hipStream_t);
void edl_sparse_ftrl(float*, float*, float*, const float*, const int32_t*,
                     int64_t, int64_t, float, float, float, float,
                     const int32_t*, hipStream_t);
void edl_dense_rmsprop(float*, float*, float*, float*, const float*, int64_t,
                       float, float, float, float, hipStream_t);
void edl_dense_adadelta(float*, float*, float*, const float*, int64_t, float,
                        float, float, hipStream_t);
void edl_dense_adamax(float*, float*, float*, const float*, int64_t, float,
                      float, float, float, hipStream_t);
void edl_dense_nadam(float*, float*, float*, const float*, int64_t, float,
                     float, float, float, float, float, float, hipStream_t);
void edl_sparse_rmsprop(float*, float*, float*, float*, const float*,
                        const int32_t*, int64_t, int64_t, float, float, float,
                        float, const int32_t*, hipStream_t);
void edl_sparse_adadelta(float*, float*, float*, const float*, const int32_t*,
                         int64_t, int64_t, float, float, float,
                         const int32_t*, hipStream_t);
void edl_sparse_adamax(float*, float*, float*, const float*, const int32_t*,
                       int64_t, int64_t, float, float, float, float,
                       const int32_t*, hipStream_t);
void edl_sparse_nadam(float*, float*, float*, const float*, const int32_t*,
                      int64_t, int64_t, float, float, float, float, float,
                      float, float, const int32_t*, hipStream_t);
void edl_ht_lookup_or_insert(int64_t*, int32_t*, int64_t, int32_t*, int32_t,
                             const int64_t*, int64_t, int32_t*, uint8_t*,
                             int32_t*, int64_t*, hipStream_t);
void edl_detect_dup_slots(const int32_t*, int64_t, int32_t*, int32_t,
                          int32_t*, hipStream_t);
void edl_ht_lookup(const int64_t*, const int32_t*, int64_t, const int64_t*,
                   int64_t, int32_t*, hipStream_t);
void edl_ht_insert_dup(int64_t*, int32_t*, int64_t, int32_t*, int32_t,
                       const int64_t*, int64_t, int32_t*, int32_t*, int64_t*,
                       hipStream_t);
void edl_batch_compact(int32_t*, int32_t*, int64_t, int32_t*, const int32_t*,
                       int64_t, int32_t*, int32_t*, hipStream_t);
void edl_accumulate_rows(const float*, const int32_t*, int64_t, int64_t,
                         float*, hipStream_t);
void edl_init_new_rows(float*, const int32_t*, const uint8_t*,
                       const int64_t*, int64_t, int64_t, uint64_t, int, float,
                       float, hipStream_t);  // is_new may be null: slot<0 skips
void edl_gather_rows(const float*, const int32_t*, int64_t, int64_t, float*,
                     hipStream_t);
void edl_scatter_rows(float*, const int32_t*, const float*, int64_t, int64_t,
                      hipStream_t);
void edl_gemm_bias_act_bf16(const void*, const void*, const float*, void*,
                            int, int, int, int, hipStream_t);
void edl_gemm256_variant_bf16(const void*, const void*, void*, int, int,
                              int, int, hipStream_t);
void edl_fused_sgd_bf16(void*, float*, float*, const void*, int64_t, float,
                        float, bool, float, float, hipStream_t);
int edl_bn_grid_for(int64_t, int64_t);
void edl_add_relu_fwd(const void*, const void*, void*, int64_t, hipStream_t);
void edl_add_relu_bwd(const void*, const void*, void*, int64_t, hipStream_t);
void edl_bn_stats(const void*, int64_t, int64_t, float*, int, float, float*,
                  float*, float*, hipStream_t);
void edl_bn_apply(const void*, void*, int64_t, int64_t, const float*,
                  const float*, const float*, const float*, bool,
                  hipStream_t);
void edl_bn_bwd_reduce(const void*, const void*, const void*, int64_t,
                       int64_t, const float*, const float*, const float*,
                       float*, int, float*, float*, float*, float*, float*,
                       hipStream_t);
void edl_bn_bwd_apply(const void*, const void*, const void*, void*, int64_t,
                      int64_t, const float*, const float*, const float*,
                      hipStream_t);
void edl_fused_adamw_bf16(void*, float*, float*, float*, const void*, int64_t,
                          float, float, float, float, float, float, float,
                          hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  // the explicit HIP accessor (torch's current stream for device 0's
  // active context) — kernels launch on the same stream torch uses
  return c10::hip::getCurrentHIPStream().stream();
}

void check_f32_cuda(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be float32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// ------------------------------ dense optimizers ------------------------
void dense_sgd(torch::Tensor p, torch::Tensor g, double lr) {
  check_f32_cuda(p, "param");
  check_f32_cuda(g, "grad");
  TORCH_CHECK(p.numel() == g.numel(), "size mismatch");
  edl_dense_sgd(p.data_ptr<float>(), g.data_ptr<float>(), p.numel(),
                static_cast<float>(lr), cur_stream());
}

void dense_momentum(torch::Tensor p, torch::Tensor vel, torch::Tensor g,
                    double lr, double mu, bool nesterov) {
  check_f32_cuda(p, "param");
  edl_dense_momentum(p.data_ptr<float>(), vel.data_ptr<float>(),
                     g.data_ptr<float>(), p.numel(), lr, mu, nesterov,
                     cur_stream());
}

void dense_adam(torch::Tensor p, torch::Tensor m, torch::Tensor v,
                c10::optional<torch::Tensor> max_sq, torch::Tensor g,
                double lr_t, double b1, double b2, double eps) {
  check_f32_cuda(p, "param");
  edl_dense_adam(p.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                 max_sq.has_value() ? max_sq->data_ptr<float>() : nullptr,
                 g.data_ptr<float>(), p.numel(), lr_t, b1, b2, eps,
                 cur_stream());
}

void dense_adagrad(torch::Tensor p, torch::Tensor m, torch::Tensor g,
                   double lr, double eps) {
  check_f32_cuda(p, "param");
  edl_dense_adagrad(p.data_ptr<float>(), m.data_ptr<float>(),
                    g.data_ptr<float>(), p.numel(), lr, eps, cur_stream());
}

void dense_ftrl(torch::Tensor p, torch::Tensor z, torch::Tensor n,
                torch::Tensor g, double alpha, double beta, double l1,
                double l2) {
  check_f32_cuda(p, "param");
  edl_dense_ftrl(p.data_ptr<float>(), z.data_ptr<float>(), n.data_ptr<float>(),
                 g.data_ptr<float>(), p.numel(), alpha, beta, l1, l2,
                 cur_stream());
}

void dense_rmsprop(torch::Tensor p, torch::Tensor ms, torch::Tensor mom,
                   c10::optional<torch::Tensor> mg, torch::Tensor g,
                   double lr, double rho, double momentum, double eps) {
  check_f32_cuda(p, "param");
  edl_dense_rmsprop(p.data_ptr<float>(), ms.data_ptr<float>(),
                    mom.data_ptr<float>(),
                    mg.has_value() ? mg->data_ptr<float>() : nullptr,
                    g.data_ptr<float>(), p.numel(), lr, rho, momentum, eps,
                    cur_stream());
}

void dense_adadelta(torch::Tensor p, torch::Tensor ag, torch::Tensor au,
                    torch::Tensor g, double lr, double rho, double eps) {
  check_f32_cuda(p, "param");
  edl_dense_adadelta(p.data_ptr<float>(), ag.data_ptr<float>(),
                     au.data_ptr<float>(), g.data_ptr<float>(), p.numel(), lr,
                     rho, eps, cur_stream());
}

void dense_adamax(torch::Tensor p, torch::Tensor m, torch::Tensor v,
                  torch::Tensor g, double lr_t, double b1, double b2,
                  double eps) {
  check_f32_cuda(p, "param");
  edl_dense_adamax(p.data_ptr<float>(), m.data_ptr<float>(),
                   v.data_ptr<float>(), g.data_ptr<float>(), p.numel(), lr_t,
                   b1, b2, eps, cur_stream());
}

void dense_nadam(torch::Tensor p, torch::Tensor m, torch::Tensor v,
                 torch::Tensor g, double lr, double c1, double c2,
                 double vcorr, double b1, double b2, double eps) {
  check_f32_cuda(p, "param");
  edl_dense_nadam(p.data_ptr<float>(), m.data_ptr<float>(),
                  v.data_ptr<float>(), g.data_ptr<float>(), p.numel(), lr, c1,
                  c2, vcorr, b1, b2, eps, cur_stream());
}

// ------------------------------ sparse optimizers -----------------------
static void check_sparse(const torch::Tensor& arena, const torch::Tensor& g,
                         const torch::Tensor& slots) {
  check_f32_cuda(arena, "arena");
  check_f32_cuda(g, "grads");
  TORCH_CHECK(slots.scalar_type() == torch::kInt32, "slots must be int32");
  TORCH_CHECK(g.dim() == 2 && arena.dim() == 2, "2-D tensors expected");
  TORCH_CHECK(g.size(1) == arena.size(1), "dim mismatch");
  TORCH_CHECK(slots.numel() == g.size(0), "slots/grads mismatch");
}

void sparse_sgd(torch::Tensor arena, torch::Tensor g, torch::Tensor slots,
                double lr, c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_sgd(arena.data_ptr<float>(), g.data_ptr<float>(),
                 slots.data_ptr<int32_t>(), g.size(0), g.size(1), lr,
                 live.has_value() ? live->data_ptr<int32_t>() : nullptr, cur_stream());
}

void sparse_momentum(torch::Tensor arena, torch::Tensor vel, torch::Tensor g,
                     torch::Tensor slots, double lr, double mu, bool nesterov,
                     c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_momentum(arena.data_ptr<float>(), vel.data_ptr<float>(),
                      g.data_ptr<float>(), slots.data_ptr<int32_t>(),
                      g.size(0), g.size(1), lr, mu, nesterov,
                      live.has_value() ? live->data_ptr<int32_t>() : nullptr, cur_stream());
}

void sparse_adam(torch::Tensor arena, torch::Tensor m, torch::Tensor v,
                 c10::optional<torch::Tensor> max_sq, torch::Tensor g,
                 torch::Tensor slots, double lr_t, double b1, double b2,
                 double eps, c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_adam(arena.data_ptr<float>(), m.data_ptr<float>(),
                  v.data_ptr<float>(),
                  max_sq.has_value() ? max_sq->data_ptr<float>() : nullptr,
                  g.data_ptr<float>(), slots.data_ptr<int32_t>(), g.size(0),
                  g.size(1), lr_t, b1, b2, eps, live.has_value() ? live->data_ptr<int32_t>() : nullptr,
                  cur_stream());
}

void sparse_adagrad(torch::Tensor arena, torch::Tensor m, torch::Tensor g,
                    torch::Tensor slots, double lr, double eps,
                    c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_adagrad(arena.data_ptr<float>(), m.data_ptr<float>(),
                     g.data_ptr<float>(), slots.data_ptr<int32_t>(), g.size(0),
                     g.size(1), lr, eps, live.has_value() ? live->data_ptr<int32_t>() : nullptr,
                     cur_stream());
}

void sparse_ftrl(torch::Tensor arena, torch::Tensor z, torch::Tensor n,
                 torch::Tensor g, torch::Tensor slots, double alpha,
                 double beta, double l1, double l2,
                 c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_ftrl(arena.data_ptr<float>(), z.data_ptr<float>(),
                  n.data_ptr<float>(), g.data_ptr<float>(),
                  slots.data_ptr<int32_t>(), g.size(0), g.size(1), alpha, beta,
                  l1, l2, live.has_value() ? live->data_ptr<int32_t>() : nullptr, cur_stream());
}

void sparse_rmsprop(torch::Tensor arena, torch::Tensor ms, torch::Tensor mom,
                    c10::optional<torch::Tensor> mg, torch::Tensor g,
                    torch::Tensor slots, double lr, double rho,
                    double momentum, double eps,
                    c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_rmsprop(arena.data_ptr<float>(), ms.data_ptr<float>(),
                     mom.data_ptr<float>(),
                     mg.has_value() ? mg->data_ptr<float>() : nullptr,
                     g.data_ptr<float>(), slots.data_ptr<int32_t>(),
                     g.size(0), g.size(1), lr, rho, momentum, eps,
                     live.has_value() ? live->data_ptr<int32_t>() : nullptr, cur_stream());
}

void sparse_adadelta(torch::Tensor arena, torch::Tensor ag, torch::Tensor au,
                     torch::Tensor g, torch::Tensor slots, double lr,
                     double rho, double eps,
                     c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_adadelta(arena.data_ptr<float>(), ag.data_ptr<float>(),
                      au.data_ptr<float>(), g.data_ptr<float>(),
                      slots.data_ptr<int32_t>(), g.size(0), g.size(1), lr,
                      rho, eps, live.has_value() ? live->data_ptr<int32_t>() : nullptr, cur_stream());
}

void sparse_adamax(torch::Tensor arena, torch::Tensor m, torch::Tensor v,
                   torch::Tensor g, torch::Tensor slots, double lr_t,
                   double b1, double b2, double eps,
                   c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_adamax(arena.data_ptr<float>(), m.data_ptr<float>(),
                    v.data_ptr<float>(), g.data_ptr<float>(),
                    slots.data_ptr<int32_t>(), g.size(0), g.size(1), lr_t, b1,
                    b2, eps, live.has_value() ? live->data_ptr<int32_t>() : nullptr, cur_stream());
}

void sparse_nadam(torch::Tensor arena, torch::Tensor m, torch::Tensor v,
                  torch::Tensor g, torch::Tensor slots, double lr, double c1,
                  double c2, double vcorr, double b1, double b2, double eps,
                  c10::optional<torch::Tensor> live) {
  check_sparse(arena, g, slots);
  edl_sparse_nadam(arena.data_ptr<float>(), m.data_ptr<float>(),
                   v.data_ptr<float>(), g.data_ptr<float>(),
                   slots.data_ptr<int32_t>(), g.size(0), g.size(1), lr, c1,
                   c2, vcorr, b1, b2, eps, live.has_value() ? live->data_ptr<int32_t>() : nullptr,
                   cur_stream());
}

// ------------------------------ hash table ------------------------------
void ht_lookup_or_insert(torch::Tensor keys, torch::Tensor vals,
                         torch::Tensor row_counter, int64_t max_rows,
                         torch::Tensor ids, torch::Tensor out_slots,
                         torch::Tensor out_is_new, torch::Tensor error_flag,
                         c10::optional<torch::Tensor> ids_by_slot) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt64);
  TORCH_CHECK((keys.numel() & (keys.numel() - 1)) == 0,
              "capacity must be a power of two");
  TORCH_CHECK(ids.scalar_type() == torch::kInt64);
  edl_ht_lookup_or_insert(
      keys.data_ptr<int64_t>(), vals.data_ptr<int32_t>(), keys.numel(),
      row_counter.data_ptr<int32_t>(), static_cast<int32_t>(max_rows),
      ids.data_ptr<int64_t>(), ids.numel(), out_slots.data_ptr<int32_t>(),
      out_is_new.data_ptr<uint8_t>(), error_flag.data_ptr<int32_t>(),
      ids_by_slot.has_value() ? ids_by_slot->data_ptr<int64_t>() : nullptr,
      cur_stream());
}

void detect_dup_slots(torch::Tensor slots, torch::Tensor mark, int64_t tag,
                      torch::Tensor dup_flag) {
  edl_detect_dup_slots(slots.data_ptr<int32_t>(), slots.numel(),
                       mark.data_ptr<int32_t>(), static_cast<int32_t>(tag),
                       dup_flag.data_ptr<int32_t>(), cur_stream());
}

void ht_insert_dup(torch::Tensor keys, torch::Tensor vals,
                   torch::Tensor row_counter, int64_t max_rows,
                   torch::Tensor ids, torch::Tensor new_slots,
                   torch::Tensor error_flag,
                   c10::optional<torch::Tensor> ids_by_slot) {
  TORCH_CHECK((keys.numel() & (keys.numel() - 1)) == 0,
              "capacity must be a power of two");
  edl_ht_insert_dup(keys.data_ptr<int64_t>(), vals.data_ptr<int32_t>(),
                    keys.numel(), row_counter.data_ptr<int32_t>(),
                    static_cast<int32_t>(max_rows), ids.data_ptr<int64_t>(),
                    ids.numel(), new_slots.data_ptr<int32_t>(),
                    error_flag.data_ptr<int32_t>(),
                    ids_by_slot.has_value() ? ids_by_slot->data_ptr<int64_t>()
                                            : nullptr,
                    cur_stream());
}

void batch_compact(torch::Tensor ht_keys, torch::Tensor ht_vals,
                   torch::Tensor counter, torch::Tensor slots,
                   torch::Tensor unique_slots, torch::Tensor compact_idx) {
  TORCH_CHECK((ht_keys.numel() & (ht_keys.numel() - 1)) == 0,
              "capacity must be a power of two");
  edl_batch_compact(ht_keys.data_ptr<int32_t>(), ht_vals.data_ptr<int32_t>(),
                    ht_keys.numel(), counter.data_ptr<int32_t>(),
                    slots.data_ptr<int32_t>(), slots.numel(),
                    unique_slots.data_ptr<int32_t>(),
                    compact_idx.data_ptr<int32_t>(), cur_stream());
}

void accumulate_rows(torch::Tensor grads, torch::Tensor compact_idx,
                     torch::Tensor acc) {
  check_f32_cuda(grads, "grads");
  check_f32_cuda(acc, "acc");
  edl_accumulate_rows(grads.data_ptr<float>(),
                      compact_idx.data_ptr<int32_t>(), grads.size(0),
                      grads.size(1), acc.data_ptr<float>(), cur_stream());
}

void ht_lookup(torch::Tensor keys, torch::Tensor vals, torch::Tensor ids,
               torch::Tensor out_slots) {
  edl_ht_lookup(keys.data_ptr<int64_t>(), vals.data_ptr<int32_t>(),
                keys.numel(), ids.data_ptr<int64_t>(), ids.numel(),
                out_slots.data_ptr<int32_t>(), cur_stream());
}

void init_new_rows(torch::Tensor arena, torch::Tensor slots,
                   c10::optional<torch::Tensor> is_new, torch::Tensor ids,
                   int64_t seed, int64_t mode, double a, double b) {
  check_f32_cuda(arena, "arena");
  TORCH_CHECK(ids.scalar_type() == torch::kInt64);
  TORCH_CHECK(mode >= 0 && mode <= 3, "init mode must be 0..3");
  edl_init_new_rows(arena.data_ptr<float>(), slots.data_ptr<int32_t>(),
                    is_new.has_value() ? is_new->data_ptr<uint8_t>() : nullptr,
                    ids.data_ptr<int64_t>(), slots.numel(), arena.size(1),
                    static_cast<uint64_t>(seed), static_cast<int>(mode), a, b,
                    cur_stream());
}

torch::Tensor gather_rows(torch::Tensor arena, torch::Tensor slots) {
  check_f32_cuda(arena, "arena");
  auto out = torch::empty({slots.numel(), arena.size(1)}, arena.options());
  edl_gather_rows(arena.data_ptr<float>(), slots.data_ptr<int32_t>(),
                  slots.numel(), arena.size(1), out.data_ptr<float>(),
                  cur_stream());
  return out;
}

void scatter_rows(torch::Tensor arena, torch::Tensor slots,
                  torch::Tensor rows) {
  check_f32_cuda(arena, "arena");
  check_f32_cuda(rows, "rows");
  edl_scatter_rows(arena.data_ptr<float>(), slots.data_ptr<int32_t>(),
                   rows.data_ptr<float>(), slots.numel(), arena.size(1),
                   cur_stream());
}

// ------------------------------ fused GEMM ------------------------------
// C[M,N] = act(A[M,K] @ B[N,K]^T + bias), bf16 in/out, f32 accumulate.
torch::Tensor gemm_bias_act(torch::Tensor a, torch::Tensor b,
                            c10::optional<torch::Tensor> bias, int64_t act) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "GPU tensors expected");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "bf16 expected");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2);
  TORCH_CHECK(a.size(1) == b.size(1), "K mismatch");
  TORCH_CHECK(a.size(1) % 64 == 0, "K must be a multiple of 64 (pad)");
  const float* bias_ptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias must be f32");
    TORCH_CHECK(bias->numel() == b.size(0));
    bias_ptr = bias->data_ptr<float>();
  }
  auto c = torch::empty({a.size(0), b.size(0)}, a.options());
  edl_gemm_bias_act_bf16(a.data_ptr(), b.data_ptr(), bias_ptr, c.data_ptr(),
                         a.size(0), b.size(0), a.size(1),
                         static_cast<int>(act), cur_stream());
  return c;
}

// benchmark-only: 256^2 kernel with explicit barrier-variant selection
torch::Tensor gemm256_bench(torch::Tensor a, torch::Tensor b, int64_t bars) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.size(1) == b.size(1) && a.size(1) % 64 == 0);
  auto c = torch::empty({a.size(0), b.size(0)}, a.options());
  edl_gemm256_variant_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                           a.size(0), b.size(0), a.size(1),
                           static_cast<int>(bars), cur_stream());
  return c;
}

// ------------------------------ batch norm ------------------------------
static void check_bn_xc(const torch::Tensor& x, const char* n) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16,
              n, " must be bf16 cuda");
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), n, " must be [R,C] contig");
  int64_t C = x.size(1);
  TORCH_CHECK(C >= 8 && C % 8 == 0 && 256 % (C / 8) == 0,
              "bn kernels need C in {8,16,...,2048} (pow2-ish); got ", C);
}

torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.numel() == b.numel() && a.numel() % 8 == 0);
  auto z = torch::empty_like(a);
  edl_add_relu_fwd(a.data_ptr(), b.data_ptr(), z.data_ptr(), a.numel(),
                   cur_stream());
  return z;
}

torch::Tensor add_relu_bwd(torch::Tensor dz, torch::Tensor z) {
  TORCH_CHECK(dz.numel() == z.numel() && dz.numel() % 8 == 0);
  auto dg = torch::empty_like(dz);
  edl_add_relu_bwd(dz.data_ptr(), z.data_ptr(), dg.data_ptr(), dz.numel(),
                   cur_stream());
  return dg;
}

// returns (mean, var_biased, rstd)
std::vector<torch::Tensor> bn_stats(torch::Tensor x, double eps) {
  check_bn_xc(x, "x");
  auto opts = x.options().dtype(torch::kFloat32);
  int64_t C = x.size(1);
  int G = edl_bn_grid_for(x.size(0), C);
  auto part = torch::empty({(int64_t)G, 2 * C}, opts);
  auto mean = torch::empty({C}, opts);
  auto var = torch::empty({C}, opts);
  auto rstd = torch::empty({C}, opts);
  edl_bn_stats(x.data_ptr(), x.size(0), C, part.data_ptr<float>(), G,
               static_cast<float>(eps), mean.data_ptr<float>(),
               var.data_ptr<float>(), rstd.data_ptr<float>(), cur_stream());
  return {mean, var, rstd};
}

torch::Tensor bn_apply(torch::Tensor x, torch::Tensor mean,
                       torch::Tensor rstd, c10::optional<torch::Tensor> gamma,
                       c10::optional<torch::Tensor> beta, bool relu) {
  check_bn_xc(x, "x");
  auto y = torch::empty_like(x);
  edl_bn_apply(x.data_ptr(), y.data_ptr(), x.size(0), x.size(1),
               mean.data_ptr<float>(), rstd.data_ptr<float>(),
               gamma.has_value() ? gamma->data_ptr<float>() : nullptr,
               beta.has_value() ? beta->data_ptr<float>() : nullptr, relu,
               cur_stream());
  return y;
}

// returns (sum_dy=dbeta, sum_dy_xhat=dgamma, a, b, c) — a/b/c are the
// dx = a*dy - b*x + c per-channel coefficients
std::vector<torch::Tensor> bn_bwd_reduce(torch::Tensor x, torch::Tensor dy,
                                         c10::optional<torch::Tensor> relu_out,
                                         torch::Tensor mean,
                                         torch::Tensor rstd,
                                         c10::optional<torch::Tensor> gamma) {
  check_bn_xc(x, "x");
  check_bn_xc(dy, "dy");
  auto opts = x.options().dtype(torch::kFloat32);
  int64_t C = x.size(1);
  int G = edl_bn_grid_for(x.size(0), C);
  auto part = torch::empty({(int64_t)G, 2 * C}, opts);
  auto s1 = torch::empty({C}, opts);
  auto s2 = torch::empty({C}, opts);
  auto ca = torch::empty({C}, opts);
  auto cb = torch::empty({C}, opts);
  auto cc = torch::empty({C}, opts);
  edl_bn_bwd_reduce(x.data_ptr(), dy.data_ptr(),
                    relu_out.has_value() ? relu_out->data_ptr() : nullptr,
                    x.size(0), C, mean.data_ptr<float>(),
                    rstd.data_ptr<float>(),
                    gamma.has_value() ? gamma->data_ptr<float>() : nullptr,
                    part.data_ptr<float>(), G, s1.data_ptr<float>(),
                    s2.data_ptr<float>(), ca.data_ptr<float>(),
                    cb.data_ptr<float>(), cc.data_ptr<float>(),
                    cur_stream());
  return {s1, s2, ca, cb, cc};
}

torch::Tensor bn_bwd_apply(torch::Tensor x, torch::Tensor dy,
                           c10::optional<torch::Tensor> relu_out,
                           torch::Tensor c1, torch::Tensor c2,
                           torch::Tensor c3) {
  check_bn_xc(x, "x");
  check_bn_xc(dy, "dy");
  auto dx = torch::empty_like(dy);
  edl_bn_bwd_apply(x.data_ptr(), dy.data_ptr(),
                   relu_out.has_value() ? relu_out->data_ptr() : nullptr,
                   dx.data_ptr(), x.size(0), x.size(1),
                   c1.data_ptr<float>(), c2.data_ptr<float>(),
                   c3.data_ptr<float>(), cur_stream());
  return dx;
}

// ---------------------- worker-side fused optimizers --------------------
void fused_sgd_bf16(torch::Tensor p, torch::Tensor master, torch::Tensor vel,
                    torch::Tensor g, double lr, double mu, bool nesterov,
                    double weight_decay, double grad_scale) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(g.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(master.scalar_type() == torch::kFloat32);
  TORCH_CHECK(p.numel() == master.numel() && p.numel() == g.numel());
  edl_fused_sgd_bf16(p.data_ptr(), master.data_ptr<float>(),
                     vel.data_ptr<float>(), g.data_ptr(), p.numel(), lr, mu,
                     nesterov, weight_decay, grad_scale, cur_stream());
}

void fused_adamw_bf16(torch::Tensor p, torch::Tensor master, torch::Tensor m,
                      torch::Tensor v, torch::Tensor g, double lr_t, double b1,
                      double b2, double eps, double weight_decay, double lr,
                      double grad_scale) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(g.scalar_type() == torch::kBFloat16);
  edl_fused_adamw_bf16(p.data_ptr(), master.data_ptr<float>(),
                       m.data_ptr<float>(), v.data_ptr<float>(), g.data_ptr(),
                       p.numel(), lr_t, b1, b2, eps, weight_decay, lr,
                       grad_scale, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "elasticdl_amd MI355X kernels";
  m.def("dense_sgd", &dense_sgd);
  m.def("dense_momentum", &dense_momentum);
  m.def("dense_adam", &dense_adam);
  m.def("dense_adagrad", &dense_adagrad);
  m.def("dense_ftrl", &dense_ftrl);
  m.def("sparse_sgd", &sparse_sgd, py::arg("arena"), py::arg("g"),
        py::arg("slots"), py::arg("lr"), py::arg("live") = py::none());
  m.def("sparse_momentum", &sparse_momentum, py::arg("arena"),
        py::arg("vel"), py::arg("g"), py::arg("slots"), py::arg("lr"),
        py::arg("mu"), py::arg("nesterov"), py::arg("live") = py::none());
  m.def("sparse_adam", &sparse_adam, py::arg("arena"), py::arg("m"),
        py::arg("v"), py::arg("max_sq"), py::arg("g"), py::arg("slots"),
        py::arg("lr_t"), py::arg("b1"), py::arg("b2"), py::arg("eps"),
        py::arg("live") = py::none());
  m.def("sparse_adagrad", &sparse_adagrad, py::arg("arena"), py::arg("m"),
        py::arg("g"), py::arg("slots"), py::arg("lr"), py::arg("eps"),
        py::arg("live") = py::none());
  m.def("sparse_ftrl", &sparse_ftrl, py::arg("arena"), py::arg("z"),
        py::arg("n"), py::arg("g"), py::arg("slots"), py::arg("alpha"),
        py::arg("beta"), py::arg("l1"), py::arg("l2"),
        py::arg("live") = py::none());
  m.def("dense_rmsprop", &dense_rmsprop);
  m.def("dense_adadelta", &dense_adadelta);
  m.def("dense_adamax", &dense_adamax);
  m.def("dense_nadam", &dense_nadam);
  m.def("sparse_rmsprop", &sparse_rmsprop, py::arg("arena"), py::arg("ms"),
        py::arg("mom"), py::arg("mg"), py::arg("g"), py::arg("slots"),
        py::arg("lr"), py::arg("rho"), py::arg("momentum"), py::arg("eps"),
        py::arg("live") = py::none());
  m.def("sparse_adadelta", &sparse_adadelta, py::arg("arena"),
        py::arg("ag"), py::arg("au"), py::arg("g"), py::arg("slots"),
        py::arg("lr"), py::arg("rho"), py::arg("eps"),
        py::arg("live") = py::none());
  m.def("sparse_adamax", &sparse_adamax, py::arg("arena"), py::arg("m"),
        py::arg("v"), py::arg("g"), py::arg("slots"), py::arg("lr_t"),
        py::arg("b1"), py::arg("b2"), py::arg("eps"),
        py::arg("live") = py::none());
  m.def("sparse_nadam", &sparse_nadam, py::arg("arena"), py::arg("m"),
        py::arg("v"), py::arg("g"), py::arg("slots"), py::arg("lr"),
        py::arg("c1"), py::arg("c2"), py::arg("vcorr"), py::arg("b1"),
        py::arg("b2"), py::arg("eps"), py::arg("live") = py::none());
  m.def("ht_lookup_or_insert", &ht_lookup_or_insert);
  m.def("ht_lookup", &ht_lookup);
  m.def("ht_insert_dup", &ht_insert_dup);
  m.def("detect_dup_slots", &detect_dup_slots);
  m.def("batch_compact", &batch_compact);
  m.def("accumulate_rows", &accumulate_rows);
  m.def("init_new_rows", &init_new_rows);
  m.def("gather_rows", &gather_rows);
  m.def("scatter_rows", &scatter_rows);
  m.def("gemm_bias_act", &gemm_bias_act);
  m.def("gemm256_bench", &gemm256_bench);
  m.def("bn_stats", &bn_stats);
  m.def("add_relu_fwd", &add_relu_fwd);
  m.def("add_relu_bwd", &add_relu_bwd);
  m.def("bn_apply", &bn_apply);
  m.def("bn_bwd_reduce", &bn_bwd_reduce);
  m.def("bn_bwd_apply", &bn_bwd_apply);
  m.def("fused_sgd_bf16", &fused_sgd_bf16);
  m.def("fused_adamw_bf16", &fused_adamw_bf16);
}
