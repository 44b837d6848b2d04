// Host-side kernel launchers (kept in HIP TU; called from the torch binding).
#include <hip/hip_runtime.h>

#include <cstdio>
#include <stdexcept>

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e) + " at " __FILE__);   \
    }                                                                      \
  } while (0)

extern "C" __global__ void k1_grad_llh(const float*, const long long*,
                                       const int*, const float*, const int*,
                                       float*, double*, int, int, float,
                                       float);
extern "C" __global__ void k4_llh_only(const float*, const long long*,
                                       const int*, const float*, const int*,
                                       double*, int, int, float, float);
extern "C" __global__ void k2_linesearch(const float*, const long long*,
                                         const int*, const float*,
                                         const float*, const double*,
                                         const int*, const float*, float*,
                                         int, int, int, float, float, float,
                                         float, float);

static void allow_large_lds(const void* func, size_t bytes) {
  if (bytes > 65536) {
    HIP_CHECK(hipFuncSetAttribute(
        func, hipFuncAttributeMaxDynamicSharedMemorySize, (int)bytes));
  }
}

extern "C" void launch_k1(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const int* order, float* grad, double* llh,
                          int n_local, int K, float min_p, float max_p,
                          hipStream_t stream) {
  if (n_local == 0) return;
  const size_t lds = (size_t)K * 4 + 4 * sizeof(double) + 8 * sizeof(float);
  allow_large_lds((const void*)k1_grad_llh, lds);
  hipLaunchKernelGGL(k1_grad_llh, dim3(n_local), dim3(256), lds, stream, F,
                     indptr, indices, sumF, order, grad, llh, n_local, K,
                     min_p, max_p);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k4(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const int* order, double* llh, int n_local, int K,
                          float min_p, float max_p, hipStream_t stream) {
  if (n_local == 0) return;
  hipLaunchKernelGGL(k4_llh_only, dim3(n_local), dim3(256), 0, stream, F,
                     indptr, indices, sumF, order, llh, n_local, K, min_p,
                     max_p);
  HIP_CHECK(hipGetLastError());
}

extern "C" void launch_k2(const float* F, const long long* indptr,
                          const int* indices, const float* sumF,
                          const float* grad, const double* llh,
                          const int* order, const float* ladder, float* best,
                          int n_local, int K, int n_ladder, float alpha,
                          float min_p, float max_p, float min_f, float max_f,
                          hipStream_t stream) {
  if (n_local == 0) return;
  if (n_ladder > 16) throw std::runtime_error("ladder length > 16 unsupported");
  hipLaunchKernelGGL(k2_linesearch, dim3(n_local), dim3(256), 0, stream, F,
                     indptr, indices, sumF, grad, llh, order, ladder, best,
                     n_local, K, n_ladder, alpha, min_p, max_p, min_f, max_f);
  HIP_CHECK(hipGetLastError());
}
