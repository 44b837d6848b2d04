// Python bindings for the BigCLAM CDNA4 kernels (bigclam._C).
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

extern "C" void launch_k1(const float*, const long long*, const int*,
                          const float*, const int*, float*, double*, int, int,
                          float, float, hipStream_t);
extern "C" void launch_k4(const float*, const long long*, const int*,
                          const float*, const int*, double*, int, int, float,
                          float, hipStream_t);
extern "C" void launch_k2(const float*, const long long*, const int*,
                          const float*, const float*, const double*,
                          const int*, const float*, float*, int, int, int,
                          float, float, float, float, float, hipStream_t);
extern "C" void launch_k3(float*, const float*, const float*, int, int,
                          float, float, hipStream_t);
extern "C" void launch_k1_bf16(const void*, const long long*, const int*,
                               const float*, const int*, float*, double*,
                               int, int, float, float, hipStream_t);
extern "C" void launch_k4_bf16(const void*, const long long*, const int*,
                               const float*, const int*, double*, int, int,
                               float, float, hipStream_t);
extern "C" void launch_k2_bf16(const void*, const long long*, const int*,
                               const float*, const float*, const double*,
                               const int*, const float*, float*, int, int,
                               int, float, float, float, float, float,
                               hipStream_t);
extern "C" void launch_k3_bf16(void*, const float*, const float*, int, int,
                               float, float, hipStream_t);
extern "C" void launch_k5(const long long*, const int*, double*, int, double,
                          hipStream_t);
extern "C" void launch_kf(const float*, const long long*, const int*,
                          const float*, const int*, float*, double*,
                          const float*, float*, int, int, int, float, float,
                          float, float, float, hipStream_t);
extern "C" void launch_kf_bf16(const void*, const long long*, const int*,
                               const float*, const int*, float*, double*,
                               const float*, float*, int, int, int, float,
                               float, float, float, float, hipStream_t);
extern "C" void launch_k3_colsum_bf16(void*, const float*, const float*,
                                      float*, int, int, float, float,
                                      hipStream_t);
extern "C" void launch_kf_mfma(const float*, const long long*, const int*,
                               const float*, const int*, float*, double*,
                               const float*, float*, int, int, int, float,
                               float, float, float, float, hipStream_t);
extern "C" void launch_kf_mfma_bf16(const void*, const long long*, const int*,
                                    const float*, const int*, float*, double*,
                                    const float*, float*, int, int, int,
                                    float, float, float, float, float,
                                    hipStream_t);
extern "C" void launch_k1_chunked(const void*, int, const long long*,
                                  const int*, const float*, const int*,
                                  float*, float*, double*, int, int, float,
                                  float, hipStream_t);
extern "C" void launch_k6(void*, int, int, const long long*,
                          const long long*, const long long*, int, long long,
                          long long, int, hipStream_t);
extern "C" void launch_kcs_lists(const void*, int, int, int,
                                 const long long*, const int*, const float*,
                                 const int*, const unsigned char*, int,
                                 float*, hipStream_t);
extern "C" void launch_k7_count(const void*, int, int, int, int, float, int*,
                                hipStream_t);
extern "C" void launch_k7_fill(const void*, int, int, int, int, float,
                               const long long*, int*, hipStream_t);
extern "C" void launch_mfma_probe_bf16(const void*, const void*, float*,
                                       hipStream_t);
extern "C" void launch_mfma_probe_f32(const float*, const float*, float*,
                                      hipStream_t);

namespace {

#define CHECK_IN(t, type)                                            \
  TORCH_CHECK(t.is_cuda(), #t " must be on GPU");                    \
  TORCH_CHECK(t.is_contiguous(), #t " must be contiguous");          \
  TORCH_CHECK(t.scalar_type() == type, #t " has wrong dtype");

#define CHECK_F(t)                                                          \
  TORCH_CHECK(t.is_cuda(), #t " must be on GPU");                           \
  TORCH_CHECK(t.is_contiguous(), #t " must be contiguous");                 \
  TORCH_CHECK(t.scalar_type() == torch::kFloat32 ||                         \
                  t.scalar_type() == torch::kBFloat16,                      \
              #t " must be fp32 or bf16");

bool is_bf16(const torch::Tensor& t) {
  return t.scalar_type() == torch::kBFloat16;
}

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void edge_grad_llh(torch::Tensor F, torch::Tensor indptr,
                   torch::Tensor indices, torch::Tensor sumF,
                   torch::Tensor order, torch::Tensor grad, torch::Tensor llh,
                   double min_p, double max_p) {
  CHECK_F(F);
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(sumF, torch::kFloat32);
  CHECK_IN(order, torch::kInt32);
  CHECK_IN(grad, torch::kFloat32);
  CHECK_IN(llh, torch::kFloat64);
  const int n_local = (int)indptr.size(0) - 1;
  // grid = #nodes listed in `order` (may be a subset: halo overlap split)
  const int n_blocks = (int)order.size(0);
  const int K = (int)F.size(1);
  TORCH_CHECK(grad.size(0) == n_local && grad.size(1) == K);
  const auto ip = reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>());
  if (is_bf16(F)) {
    TORCH_CHECK(K % 8 == 0, "bf16 K must be padded to a multiple of 8");
    launch_k1_bf16(F.data_ptr(), ip, indices.data_ptr<int>(),
                   sumF.data_ptr<float>(), order.data_ptr<int>(),
                   grad.data_ptr<float>(), llh.data_ptr<double>(), n_blocks,
                   K, (float)min_p, (float)max_p, current_stream());
  } else {
    TORCH_CHECK(K % 4 == 0, "K must be padded to a multiple of 4");
    launch_k1(F.data_ptr<float>(), ip, indices.data_ptr<int>(),
              sumF.data_ptr<float>(), order.data_ptr<int>(),
              grad.data_ptr<float>(), llh.data_ptr<double>(), n_blocks, K,
              (float)min_p, (float)max_p, current_stream());
  }
}

void llh_only(torch::Tensor F, torch::Tensor indptr, torch::Tensor indices,
              torch::Tensor sumF, torch::Tensor order, torch::Tensor llh,
              double min_p, double max_p) {
  CHECK_F(F);
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(sumF, torch::kFloat32);
  CHECK_IN(order, torch::kInt32);
  CHECK_IN(llh, torch::kFloat64);
  const int n_local = (int)indptr.size(0) - 1;
  const int K = (int)F.size(1);
  const auto ip = reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>());
  if (is_bf16(F)) {
    TORCH_CHECK(K % 8 == 0, "bf16 K must be padded to a multiple of 8");
    launch_k4_bf16(F.data_ptr(), ip, indices.data_ptr<int>(),
                   sumF.data_ptr<float>(), order.data_ptr<int>(),
                   llh.data_ptr<double>(), n_local, K, (float)min_p,
                   (float)max_p, current_stream());
  } else {
    TORCH_CHECK(K % 4 == 0, "K must be padded to a multiple of 4");
    launch_k4(F.data_ptr<float>(), ip, indices.data_ptr<int>(),
              sumF.data_ptr<float>(), order.data_ptr<int>(),
              llh.data_ptr<double>(), n_local, K, (float)min_p, (float)max_p,
              current_stream());
  }
}

void linesearch(torch::Tensor F, torch::Tensor indptr, torch::Tensor indices,
                torch::Tensor sumF, torch::Tensor grad, torch::Tensor llh,
                torch::Tensor order, torch::Tensor ladder, torch::Tensor best,
                double alpha, double min_p, double max_p, double min_f,
                double max_f) {
  CHECK_F(F);
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(sumF, torch::kFloat32);
  CHECK_IN(grad, torch::kFloat32);
  CHECK_IN(llh, torch::kFloat64);
  CHECK_IN(order, torch::kInt32);
  CHECK_IN(ladder, torch::kFloat32);
  CHECK_IN(best, torch::kFloat32);
  // grid = #nodes listed in `order` (may be a subset: the sparse path's
  // dense hub remainder); unlisted best[] entries stay untouched
  const int n_blocks = (int)order.size(0);
  const int K = (int)F.size(1);
  const auto ip = reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>());
  if (is_bf16(F)) {
    TORCH_CHECK(K % 8 == 0, "bf16 K must be padded to a multiple of 8");
    launch_k2_bf16(F.data_ptr(), ip, indices.data_ptr<int>(),
                   sumF.data_ptr<float>(), grad.data_ptr<float>(),
                   llh.data_ptr<double>(), order.data_ptr<int>(),
                   ladder.data_ptr<float>(), best.data_ptr<float>(),
                   n_blocks, K, (int)ladder.size(0), (float)alpha,
                   (float)min_p, (float)max_p, (float)min_f, (float)max_f,
                   current_stream());
  } else {
    TORCH_CHECK(K % 4 == 0, "K must be padded to a multiple of 4");
    launch_k2(F.data_ptr<float>(), ip, indices.data_ptr<int>(),
              sumF.data_ptr<float>(), grad.data_ptr<float>(),
              llh.data_ptr<double>(), order.data_ptr<int>(),
              ladder.data_ptr<float>(), best.data_ptr<float>(), n_blocks, K,
              (int)ladder.size(0), (float)alpha, (float)min_p, (float)max_p,
              (float)min_f, (float)max_f, current_stream());
  }
}

void apply_step(torch::Tensor F_local, torch::Tensor grad,
                torch::Tensor steps, double min_f, double max_f) {
  CHECK_F(F_local);
  CHECK_IN(grad, torch::kFloat32);
  CHECK_IN(steps, torch::kFloat32);
  const int n_local = (int)F_local.size(0);
  const int K = (int)F_local.size(1);
  TORCH_CHECK(grad.size(0) == n_local && grad.size(1) == K);
  TORCH_CHECK(steps.size(0) == n_local);
  if (is_bf16(F_local)) {
    TORCH_CHECK(K % 8 == 0, "bf16 K must be padded to a multiple of 8");
    launch_k3_bf16(F_local.data_ptr(), grad.data_ptr<float>(),
                   steps.data_ptr<float>(), n_local, K, (float)min_f,
                   (float)max_f, current_stream());
  } else {
    TORCH_CHECK(K % 4 == 0, "K must be padded to a multiple of 4");
    launch_k3(F_local.data_ptr<float>(), grad.data_ptr<float>(),
              steps.data_ptr<float>(), n_local, K, (float)min_f,
              (float)max_f, current_stream());
  }
}

// n_mfma: the first n_mfma entries of `order` (degree-descending, so the
// high-degree prefix) run the MFMA phase-B kernel; the rest the direct
// one.  The two launches cover disjoint nodes.
void fused_grad_ls(torch::Tensor F, torch::Tensor indptr,
                   torch::Tensor indices, torch::Tensor sumF,
                   torch::Tensor order, torch::Tensor grad, torch::Tensor llh,
                   torch::Tensor ladder, torch::Tensor best, double alpha,
                   double min_p, double max_p, double min_f, double max_f,
                   int64_t n_mfma) {
  CHECK_F(F);
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(sumF, torch::kFloat32);
  CHECK_IN(order, torch::kInt32);
  CHECK_IN(grad, torch::kFloat32);
  CHECK_IN(llh, torch::kFloat64);
  CHECK_IN(ladder, torch::kFloat32);
  CHECK_IN(best, torch::kFloat32);
  const int n_local = (int)indptr.size(0) - 1;
  const int n_blocks = (int)order.size(0);
  const int K = (int)F.size(1);
  TORCH_CHECK(grad.size(0) == n_local && grad.size(1) == K);
  const auto ip =
      reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>());
  TORCH_CHECK(n_mfma >= 0 && n_mfma <= n_blocks, "bad n_mfma");
  const int n_hi = (int)n_mfma;
  const int n_lo = n_blocks - n_hi;
  const int* ord = order.data_ptr<int>();
  if (is_bf16(F)) {
    TORCH_CHECK(K % 8 == 0 && K <= 26000,
                "bf16 fused kernel: K padded to 8, K <= 26000 (the direct "
                "kernel additionally requires K <= 16384 for its share)");
    launch_kf_mfma_bf16(F.data_ptr(), ip, indices.data_ptr<int>(),
                        sumF.data_ptr<float>(), ord, grad.data_ptr<float>(),
                        llh.data_ptr<double>(), ladder.data_ptr<float>(),
                        best.data_ptr<float>(), n_hi, K, (int)ladder.size(0),
                        (float)alpha, (float)min_p, (float)max_p,
                        (float)min_f, (float)max_f, current_stream());
    launch_kf_bf16(F.data_ptr(), ip, indices.data_ptr<int>(),
                   sumF.data_ptr<float>(), ord + n_hi,
                   grad.data_ptr<float>(), llh.data_ptr<double>(),
                   ladder.data_ptr<float>(), best.data_ptr<float>(), n_lo,
                   K, (int)ladder.size(0), (float)alpha, (float)min_p,
                   (float)max_p, (float)min_f, (float)max_f,
                   current_stream());
  } else {
    TORCH_CHECK(K % 4 == 0 && K <= 8192, "fused kernel: K padded, K <= 8192");
    launch_kf_mfma(F.data_ptr<float>(), ip, indices.data_ptr<int>(),
                   sumF.data_ptr<float>(), ord, grad.data_ptr<float>(),
                   llh.data_ptr<double>(), ladder.data_ptr<float>(),
                   best.data_ptr<float>(), n_hi, K, (int)ladder.size(0),
                   (float)alpha, (float)min_p, (float)max_p, (float)min_f,
                   (float)max_f, current_stream());
    launch_kf(F.data_ptr<float>(), ip, indices.data_ptr<int>(),
              sumF.data_ptr<float>(), ord + n_hi, grad.data_ptr<float>(),
              llh.data_ptr<double>(), ladder.data_ptr<float>(),
              best.data_ptr<float>(), n_lo, K, (int)ladder.size(0),
              (float)alpha, (float)min_p, (float)max_p, (float)min_f,
              (float)max_f, current_stream());
  }
}

// K3 + column partial sums fused (bf16 storage): commits F in place and
// writes per-stripe fp32 column partials; sumF = partials.sum(0) on the
// caller side (deterministic stage-2).
void apply_step_colsum(torch::Tensor F_local, torch::Tensor grad,
                       torch::Tensor steps, torch::Tensor partials,
                       double min_f, double max_f) {
  TORCH_CHECK(F_local.scalar_type() == torch::kBFloat16 &&
                  F_local.is_contiguous(),
              "apply_step_colsum is the bf16 path");
  CHECK_IN(grad, torch::kFloat32);
  CHECK_IN(steps, torch::kFloat32);
  CHECK_IN(partials, torch::kFloat32);
  const int n_local = (int)F_local.size(0);
  const int K = (int)F_local.size(1);
  TORCH_CHECK(K % 8 == 0, "bf16 K must be padded to a multiple of 8");
  TORCH_CHECK(grad.size(0) == n_local && grad.size(1) == K);
  TORCH_CHECK(steps.size(0) == n_local);
  TORCH_CHECK(partials.size(0) == (n_local + 511) / 512 &&
              partials.size(1) == K);
  launch_k3_colsum_bf16(F_local.data_ptr(), grad.data_ptr<float>(),
                        steps.data_ptr<float>(), partials.data_ptr<float>(),
                        n_local, K, (float)min_f, (float)max_f,
                        current_stream());
}

// On-device MFMA C/D-layout probe: D = A @ B with Bc = B column-major.
void mfma_probe(torch::Tensor A, torch::Tensor Bc, torch::Tensor D) {
  TORCH_CHECK(D.scalar_type() == torch::kFloat32 && D.is_contiguous());
  TORCH_CHECK(A.is_contiguous() && Bc.is_contiguous());
  TORCH_CHECK(A.scalar_type() == Bc.scalar_type());
  if (is_bf16(A)) {
    TORCH_CHECK(A.size(0) == 16 && A.size(1) == 32);
    launch_mfma_probe_bf16(A.data_ptr(), Bc.data_ptr(),
                           D.data_ptr<float>(), current_stream());
  } else {
    TORCH_CHECK(A.size(0) == 16 && A.size(1) == 4);
    launch_mfma_probe_f32(A.data_ptr<float>(), Bc.data_ptr<float>(),
                          D.data_ptr<float>(), current_stream());
  }
}

// Chunked large-K K1 (KD dot pass into the per-edge x buffer, then KW
// chunked weighted accumulate) — no K cap; see kd_dot_t/kw_grad_t.
void edge_grad_llh_chunked(torch::Tensor F, torch::Tensor indptr,
                           torch::Tensor indices, torch::Tensor sumF,
                           torch::Tensor order, torch::Tensor grad,
                           torch::Tensor llh, torch::Tensor xbuf,
                           double min_p, double max_p) {
  CHECK_F(F);
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(sumF, torch::kFloat32);
  CHECK_IN(order, torch::kInt32);
  CHECK_IN(grad, torch::kFloat32);
  CHECK_IN(llh, torch::kFloat64);
  CHECK_IN(xbuf, torch::kFloat32);
  const int n_local = (int)indptr.size(0) - 1;
  const int n_blocks = (int)order.size(0);
  const int K = (int)F.size(1);
  TORCH_CHECK(grad.size(0) == n_local && grad.size(1) == K);
  TORCH_CHECK(xbuf.size(0) >= indices.size(0), "xbuf too small");
  TORCH_CHECK(K % (is_bf16(F) ? 8 : 4) == 0, "K must be padded");
  const auto ip =
      reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>());
  launch_k1_chunked(F.data_ptr(), is_bf16(F) ? 1 : 0, ip,
                    indices.data_ptr<int>(), sumF.data_ptr<float>(),
                    order.data_ptr<int>(), xbuf.data_ptr<float>(),
                    grad.data_ptr<float>(), llh.data_ptr<double>(), n_blocks,
                    K, (float)min_p, (float)max_p, current_stream());
}

// Sparse-adaptive sweep (KAF/K1S/K2S/K3S — docs/sparse_sweep_design.md).
extern "C" void launch_kaf(const void*, int, int, int, int,
                           const long long*, int*, int*, float*, int,
                           const unsigned char*, hipStream_t);
extern "C" void launch_kfs(const void*, int, const long long*, const int*,
                           const float*, const int*, int, const long long*,
                           const int*, const float*, const int*,
                           const long long*, const long long*, int*, float*,
                           int*, double*, const float*, const float*,
                           float*, int, int, int, float, float, float,
                           float, float, hipStream_t);
extern "C" void launch_k3s(void*, int, const int*, int, const long long*,
                           const int*, const float*, const int*,
                           const float*, const long long*, int*, float*,
                           int*, int, int, float, float, hipStream_t);

static const long long* i64p(const torch::Tensor& t) {
  return reinterpret_cast<const long long*>(t.data_ptr<int64_t>());
}

void sparse_support(torch::Tensor F, torch::Tensor soffset,
                    torch::Tensor scount, torch::Tensor sidx,
                    torch::Tensor sval, int64_t cap, bool fill,
                    torch::Tensor dirty) {
  CHECK_F(F);
  CHECK_IN(soffset, torch::kInt64);
  CHECK_IN(scount, torch::kInt32);
  CHECK_IN(sidx, torch::kInt32);
  CHECK_IN(sval, torch::kFloat32);
  const int n_rows = (int)F.size(0);
  const int K = (int)F.size(1);
  TORCH_CHECK(scount.size(0) == n_rows && soffset.size(0) == n_rows);
  const unsigned char* dp = nullptr;
  if (dirty.numel()) {
    TORCH_CHECK(dirty.scalar_type() == torch::kUInt8 &&
                dirty.is_contiguous() && dirty.size(0) == n_rows);
    dp = dirty.data_ptr<uint8_t>();
  }
  launch_kaf(F.data_ptr(), is_bf16(F) ? 1 : 0, n_rows, K, (int)cap,
             i64p(soffset), scount.data_ptr<int>(), sidx.data_ptr<int>(),
             sval.data_ptr<float>(), fill ? 1 : 0, dp, current_stream());
}

void sparse_fused(torch::Tensor F, torch::Tensor indptr,
                  torch::Tensor indices, torch::Tensor sumF,
                  torch::Tensor order, torch::Tensor soffset,
                  torch::Tensor sidx, torch::Tensor sval,
                  torch::Tensor scount, torch::Tensor epos,
                  torch::Tensor goffset, torch::Tensor gidx,
                  torch::Tensor gval, torch::Tensor gcount,
                  torch::Tensor llh, torch::Tensor GG, torch::Tensor ladder,
                  torch::Tensor best, int64_t cap, double alpha,
                  double min_p, double max_p, double min_f, double max_f) {
  CHECK_F(F);
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(sumF, torch::kFloat32);
  CHECK_IN(order, torch::kInt32);
  CHECK_IN(soffset, torch::kInt64);
  CHECK_IN(sidx, torch::kInt32);
  CHECK_IN(sval, torch::kFloat32);
  CHECK_IN(scount, torch::kInt32);
  CHECK_IN(epos, torch::kInt64);
  CHECK_IN(goffset, torch::kInt64);
  CHECK_IN(gidx, torch::kInt32);
  CHECK_IN(gval, torch::kFloat32);
  CHECK_IN(gcount, torch::kInt32);
  CHECK_IN(llh, torch::kFloat64);
  CHECK_IN(GG, torch::kFloat32);
  CHECK_IN(ladder, torch::kFloat32);
  CHECK_IN(best, torch::kFloat32);
  const int n_blocks = (int)order.size(0);
  TORCH_CHECK(goffset.size(0) >= n_blocks && gcount.size(0) >= n_blocks);
  TORCH_CHECK(epos.size(0) == indices.size(0) + 1, "epos must be nnz+1");
  launch_kfs(F.data_ptr(), is_bf16(F) ? 1 : 0,
             reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>()),
             indices.data_ptr<int>(), sumF.data_ptr<float>(),
             order.data_ptr<int>(), n_blocks,
             reinterpret_cast<const long long*>(soffset.data_ptr<int64_t>()),
             sidx.data_ptr<int>(), sval.data_ptr<float>(),
             scount.data_ptr<int>(),
             reinterpret_cast<const long long*>(epos.data_ptr<int64_t>()),
             reinterpret_cast<const long long*>(goffset.data_ptr<int64_t>()),
             gidx.data_ptr<int>(), gval.data_ptr<float>(),
             gcount.data_ptr<int>(), llh.data_ptr<double>(),
             GG.data_ptr<float>(), ladder.data_ptr<float>(),
             best.data_ptr<float>(), (int)ladder.size(0), (int)cap,
             (int)F.size(1), (float)alpha, (float)min_p, (float)max_p,
             (float)min_f, (float)max_f, current_stream());
}

void sparse_commit(torch::Tensor F, torch::Tensor order,
                   torch::Tensor goffset, torch::Tensor gidx,
                   torch::Tensor gval, torch::Tensor gcount,
                   torch::Tensor best, torch::Tensor soffset,
                   torch::Tensor sidx, torch::Tensor sval,
                   torch::Tensor scount, int64_t cap, double min_f,
                   double max_f) {
  CHECK_F(F);
  CHECK_IN(best, torch::kFloat32);
  CHECK_IN(soffset, torch::kInt64);
  CHECK_IN(sidx, torch::kInt32);
  CHECK_IN(sval, torch::kFloat32);
  CHECK_IN(scount, torch::kInt32);
  launch_k3s(F.data_ptr(), is_bf16(F) ? 1 : 0, order.data_ptr<int>(),
             (int)order.size(0), i64p(goffset), gidx.data_ptr<int>(),
             gval.data_ptr<float>(), gcount.data_ptr<int>(),
             best.data_ptr<float>(), i64p(soffset), sidx.data_ptr<int>(),
             sval.data_ptr<float>(), scount.data_ptr<int>(), (int)cap,
             (int)F.size(1), (float)min_f, (float)max_f, current_stream());
}

// K6: seed-init F scatter from the seeds' compact adjacency (see
// k6_seed_init_t); F_local is the rank's [n_local, kp] slice.
void seed_init(torch::Tensor F_local, torch::Tensor sindptr,
               torch::Tensor snbrs, torch::Tensor seeds, int64_t start,
               int64_t stop, bool include_seed) {
  CHECK_F(F_local);
  CHECK_IN(sindptr, torch::kInt64);
  CHECK_IN(snbrs, torch::kInt64);
  CHECK_IN(seeds, torch::kInt64);
  const int n_seeds = (int)seeds.size(0);
  TORCH_CHECK(sindptr.size(0) == n_seeds + 1);
  TORCH_CHECK(stop - start == F_local.size(0));
  launch_k6(F_local.data_ptr(), is_bf16(F_local) ? 1 : 0,
            (int)F_local.size(1),
            reinterpret_cast<const long long*>(sindptr.data_ptr<int64_t>()),
            reinterpret_cast<const long long*>(snbrs.data_ptr<int64_t>()),
            reinterpret_cast<const long long*>(seeds.data_ptr<int64_t>()),
            n_seeds, (long long)start, (long long)stop,
            include_seed ? 1 : 0, current_stream());
}

// List-based column sums for the sparse path (see kcs_lists_t).
void sparse_colsum(torch::Tensor F_local, torch::Tensor soffset,
                   torch::Tensor sidx, torch::Tensor sval,
                   torch::Tensor scount, torch::Tensor dirty, int64_t cap,
                   torch::Tensor partials) {
  CHECK_F(F_local);
  CHECK_IN(soffset, torch::kInt64);
  CHECK_IN(sidx, torch::kInt32);
  CHECK_IN(sval, torch::kFloat32);
  CHECK_IN(scount, torch::kInt32);
  CHECK_IN(partials, torch::kFloat32);
  TORCH_CHECK(dirty.scalar_type() == torch::kUInt8 && dirty.is_contiguous());
  const int n_local = (int)F_local.size(0);
  const int K = (int)F_local.size(1);
  TORCH_CHECK(partials.size(0) == (n_local + 511) / 512 &&
              partials.size(1) == K);
  launch_kcs_lists(F_local.data_ptr(), is_bf16(F_local) ? 1 : 0, n_local, K,
                   i64p(soffset), sidx.data_ptr<int>(),
                   sval.data_ptr<float>(), scount.data_ptr<int>(),
                   dirty.data_ptr<uint8_t>(), (int)cap,
                   partials.data_ptr<float>(), current_stream());
}

// K7 community extraction: two deterministic passes (count, then fill
// after a host/torch prefix-sum) — see k7_membership in the .hip file.
void extract_count(torch::Tensor F_local, int64_t k_true, double delta,
                   torch::Tensor counts) {
  CHECK_F(F_local);
  CHECK_IN(counts, torch::kInt32);
  const int n = (int)F_local.size(0);
  const int ldF = (int)F_local.size(1);
  TORCH_CHECK(counts.size(0) == n);
  TORCH_CHECK(k_true >= 1 && k_true <= ldF, "bad k_true");
  launch_k7_count(F_local.data_ptr(), is_bf16(F_local) ? 1 : 0, n,
                  (int)k_true, ldF, (float)delta, counts.data_ptr<int>(),
                  current_stream());
}

void extract_fill(torch::Tensor F_local, int64_t k_true, double delta,
                  torch::Tensor offsets, torch::Tensor comms) {
  CHECK_F(F_local);
  CHECK_IN(offsets, torch::kInt64);
  CHECK_IN(comms, torch::kInt32);
  const int n = (int)F_local.size(0);
  const int ldF = (int)F_local.size(1);
  TORCH_CHECK(offsets.size(0) == n);
  TORCH_CHECK(k_true >= 1 && k_true <= ldF, "bad k_true");
  launch_k7_fill(F_local.data_ptr(), is_bf16(F_local) ? 1 : 0, n,
                 (int)k_true, ldF, (float)delta,
                 reinterpret_cast<const long long*>(
                     offsets.data_ptr<int64_t>()),
                 comms.data_ptr<int>(), current_stream());
}

void conductance(torch::Tensor indptr, torch::Tensor indices,
                 torch::Tensor cond, double total_degree) {
  CHECK_IN(indptr, torch::kInt64);
  CHECK_IN(indices, torch::kInt32);
  CHECK_IN(cond, torch::kFloat64);
  const int n = (int)indptr.size(0) - 1;
  TORCH_CHECK(cond.size(0) == n);
  launch_k5(reinterpret_cast<const long long*>(indptr.data_ptr<int64_t>()),
            indices.data_ptr<int>(), cond.data_ptr<double>(), n,
            total_degree, current_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("edge_grad_llh", &edge_grad_llh,
        "K1: fused per-node gradient + local LLH (CDNA4)");
  m.def("llh_only", &llh_only, "K4: per-node local LLH (CDNA4)");
  m.def("linesearch", &linesearch,
        "K2: 16-candidate Armijo line search in one edge pass (CDNA4)");
  m.def("apply_step", &apply_step,
        "K3: in-place projected commit F += s*grad (CDNA4)");
  m.def("conductance", &conductance,
        "K5: ego-net conductance per node (CDNA4)");
  m.def("fused_grad_ls", &fused_grad_ls,
        "KF: fused K1 gradient+LLH and K2 line search, one pass per node "
        "(MFMA tiles for the high-degree prefix)");
  m.def("apply_step_colsum", &apply_step_colsum,
        "K3+colsum fused (bf16): commit F and emit per-stripe column sums");
  m.def("sparse_support", &sparse_support,
        "KAF: per-row support compaction of F (count or fill pass)");
  m.def("sparse_fused", &sparse_fused,
        "KFS: fused compact gradient + 16-candidate Armijo for routed "
        "nodes (LDS bitmap active sets)");
  m.def("sparse_colsum", &sparse_colsum,
        "list-based column sums (sparse path; stale/over-cap rows dense)");
  m.def("sparse_commit", &sparse_commit,
        "K3S: sparse projected commit confined to the active set");
  m.def("edge_grad_llh_chunked", &edge_grad_llh_chunked,
        "K1 large-K: KD per-edge dots + KW chunked weighted accumulate");
  m.def("seed_init", &seed_init,
        "K6: seed-init F scatter (community c = neighbors of seed c)");
  m.def("extract_count", &extract_count,
        "K7 pass 1: per-row membership counts (threshold/argmax-fallback)");
  m.def("extract_fill", &extract_fill,
        "K7 pass 2: write ascending community ids at per-row offsets");
  m.def("mfma_probe", &mfma_probe,
        "MFMA C/D layout probe: one 16x16 D = A @ Bc^T tile");
}
