// Times the REAL shipping kernels (launch_k1/launch_k2/launch_k3) on the
// real bench CSR (gpurun_out_graph.bin dumped by bigclam.io.shaped_graph).
// Build: hipcc --offload-arch=gfx950 -O3 kbench.hip bigclam_kernels.hip -o kbench
// Run:   ./kbench [K]
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

extern "C" void launch_k1(const float*, const long long*, const int*,
                          const float*, const int*, float*, double*, int, int,
                          float, float, hipStream_t);
extern "C" void launch_k2(const float*, const long long*, const int*,
                          const float*, const float*, const double*,
                          const int*, const float*, float*, int, int, int,
                          float, float, float, float, float, hipStream_t);
extern "C" void launch_k3(float*, const float*, const float*, int, int, float,
                          float, hipStream_t);
extern "C" void launch_k4(const float*, const long long*, const int*,
                          const float*, const int*, double*, int, int, float,
                          float, hipStream_t);
extern "C" void launch_k1_bf16(const void*, const long long*, const int*,
                               const float*, const int*, float*, double*,
                               int, int, float, float, hipStream_t);
extern "C" void launch_k2_bf16(const void*, const long long*, const int*,
                               const float*, const float*, const double*,
                               const int*, const float*, float*, int, int,
                               int, float, float, float, float, float,
                               hipStream_t);
extern "C" void launch_k3_bf16(void*, const float*, const float*, int, int,
                               float, float, hipStream_t);
extern "C" void launch_kf(const float*, const long long*, const int*,
                          const float*, const int*, float*, double*,
                          const float*, float*, int, int, int, float, float,
                          float, float, float, hipStream_t);
extern "C" void launch_kf_mfma(const float*, const long long*, const int*,
                               const float*, const int*, float*, double*,
                               const float*, float*, int, int, int, float,
                               float, float, float, float, hipStream_t);
extern "C" void launch_kf_bf16(const void*, const long long*, const int*,
                               const float*, const int*, float*, double*,
                               const float*, float*, int, int, int, float,
                               float, float, float, float, hipStream_t);
extern "C" void launch_kf_mfma_bf16(const void*, const long long*, const int*,
                                    const float*, const int*, float*, double*,
                                    const float*, float*, int, int, int,
                                    float, float, float, float, float,
                                    hipStream_t);

#define HIP_CHECK(x)                                                \
  do {                                                              \
    hipError_t e = (x);                                             \
    if (e != hipSuccess) {                                          \
      printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                      \
    }                                                               \
  } while (0)

int main(int argc, char** argv) {
  int K = argc > 1 ? atoi(argv[1]) : 5000;
  // mode "kfb0": ONLY the bf16 all-MFMA fused timing (for rocprofv3
  // --pmc runs, where every extra dispatch multiplies collection passes)
  const bool only_kfb0 = argc > 2 && strcmp(argv[2], "kfb0") == 0;

  FILE* fp = fopen("gpurun_out_graph.bin", "rb");
  if (!fp) {
    printf("missing gpurun_out_graph.bin\n");
    return 1;
  }
  long long hdr[2];
  if (fread(hdr, 8, 2, fp) != 2) return 1;
  int N = (int)hdr[0];
  long long E = hdr[1];
  std::vector<long long> indptr(N + 1);
  std::vector<int> indices(E);
  if (fread(indptr.data(), 8, N + 1, fp) != (size_t)N + 1) return 1;
  if (fread(indices.data(), 4, E, fp) != (size_t)E) return 1;
  fclose(fp);

  std::vector<int> order(N);
  for (int i = 0; i < N; ++i) order[i] = i;
  std::sort(order.begin(), order.end(), [&](int a, int b) {
    return indptr[a + 1] - indptr[a] > indptr[b + 1] - indptr[b];
  });

  float *dF, *dG, *dSum, *dBest, *dLad;
  double* dLlh;
  long long* dIp;
  int *dIdx, *dOrd;
  HIP_CHECK(hipMalloc(&dF, (size_t)N * K * 4));
  HIP_CHECK(hipMalloc(&dG, (size_t)N * K * 4));
  HIP_CHECK(hipMalloc(&dSum, (size_t)K * 4));
  HIP_CHECK(hipMalloc(&dBest, (size_t)N * 4));
  HIP_CHECK(hipMalloc(&dLad, 16 * 4));
  HIP_CHECK(hipMalloc(&dLlh, (size_t)N * 8));
  HIP_CHECK(hipMalloc(&dIp, (N + 1) * 8));
  HIP_CHECK(hipMalloc(&dIdx, E * 4));
  HIP_CHECK(hipMalloc(&dOrd, N * 4));
  HIP_CHECK(hipMemcpy(dIp, indptr.data(), (N + 1) * 8, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dIdx, indices.data(), E * 4, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dOrd, order.data(), N * 4, hipMemcpyHostToDevice));

  // random-ish init: F in [0,1), sumF = N*0.5, grad ~ -sumF (realistic)
  std::vector<float> hF((size_t)16384);
  srand(7);
  for (auto& v : hF) v = (float)rand() / RAND_MAX;
  for (size_t off = 0; off < (size_t)N * K; off += hF.size()) {
    size_t n = std::min(hF.size(), (size_t)N * K - off);
    HIP_CHECK(hipMemcpy(dF + off, hF.data(), n * 4, hipMemcpyHostToDevice));
  }
  std::vector<float> hSum(K, N * 0.5f);
  HIP_CHECK(hipMemcpy(dSum, hSum.data(), K * 4, hipMemcpyHostToDevice));
  float hLad[16];
  for (int i = 0; i < 16; ++i) hLad[i] = powf(0.1f, (float)i);
  HIP_CHECK(hipMemcpy(dLad, hLad, 64, hipMemcpyHostToDevice));

  printf("N=%d E=%lld K=%d\n", N, E, K);

  auto time3 = [&](const char* name, auto fn) {
    fn();  // warmup (also fills grad for k2)
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t a, b;
    HIP_CHECK(hipEventCreate(&a));
    HIP_CHECK(hipEventCreate(&b));
    HIP_CHECK(hipEventRecord(a));
    for (int r = 0; r < 3; ++r) fn();
    HIP_CHECK(hipEventRecord(b));
    HIP_CHECK(hipDeviceSynchronize());
    float ms;
    HIP_CHECK(hipEventElapsedTime(&ms, a, b));
    printf("%-6s %8.2f ms\n", name, ms / 3);
    fflush(stdout);
  };

  if (!only_kfb0) time3("k1", [&] {
    launch_k1(dF, dIp, dIdx, dSum, dOrd, dG, dLlh, N, K, 1e-4f, 0.9999f, 0);
  });
  if (!only_kfb0) time3("k4", [&] {
    launch_k4(dF, dIp, dIdx, dSum, dOrd, dLlh, N, K, 1e-4f, 0.9999f, 0);
  });
  if (!only_kfb0) time3("k2", [&] {
    launch_k2(dF, dIp, dIdx, dSum, dG, dLlh, dOrd, dLad, dBest, N, K, 16,
              0.05f, 1e-4f, 0.9999f, 0.f, 1000.f, 0);
  });
  if (!only_kfb0) time3("k3", [&] {
    launch_k3(dF, dG, dBest, N, K, 0.f, 1000.f, 0);
  });

  // fused K1+K2 with the MFMA phase-B prefix at several degree splits
  auto split_at = [&](long long thr) {
    int n = 0;
    while (n < N && indptr[order[n] + 1] - indptr[order[n]] >= thr) ++n;
    return n;
  };
  auto kf_split = [&](int n_hi) {
    launch_kf_mfma(dF, dIp, dIdx, dSum, dOrd, dG, dLlh, dLad, dBest, n_hi, K,
                   16, 0.05f, 1e-4f, 0.9999f, 0.f, 1000.f, 0);
    launch_kf(dF, dIp, dIdx, dSum, dOrd + n_hi, dG, dLlh, dLad, dBest,
              N - n_hi, K, 16, 0.05f, 1e-4f, 0.9999f, 0.f, 1000.f, 0);
  };
  const int thrs[] = {0, 8, 16, 32, 64, 1 << 30};
  if (K <= 8192 && !only_kfb0) {  // fp32 fused-kernel K cap
    for (int thr : thrs) {
      const int n_hi = thr == 0 ? N : (thr == (1 << 30) ? 0 : split_at(thr));
      char name[32];
      snprintf(name, sizeof name, "kf@%-4d", thr == (1 << 30) ? -1 : thr);
      printf("  (n_mfma=%d)\n", n_hi);
      time3(name, [&] { kf_split(n_hi); });
    }
  }

  // bf16 storage path (K padded to 8 assumed by the callers; K=5000 ok
  // since kernels only need K%8==0 for uint4 rows — pad here)
  if (K % 8 == 0 || true) {
    const int Kb = (K + 7) & ~7;
    unsigned int* dFb;
    HIP_CHECK(hipMalloc(&dFb, (size_t)N * Kb * 2));
    // pack dF (fp32) into bf16 pairs on host once (slow but simple)
    {
      std::vector<float> row(Kb, 0.f);
      std::vector<unsigned int> rowb(Kb / 2);
      std::vector<unsigned int> all((size_t)N * Kb / 2);
      std::vector<float> hFall((size_t)N * K);
      HIP_CHECK(hipMemcpy(hFall.data(), dF, (size_t)N * K * 4,
                          hipMemcpyDeviceToHost));
      for (int u = 0; u < N; ++u) {
        for (int k = 0; k < K; ++k) row[k] = hFall[(size_t)u * K + k];
        for (int k = K; k < Kb; ++k) row[k] = 0.f;
        for (int k = 0; k < Kb; k += 2) {
          unsigned int l, h;
          memcpy(&l, &row[k], 4);
          memcpy(&h, &row[k + 1], 4);
          l += 0x7fffu + ((l >> 16) & 1u);
          h += 0x7fffu + ((h >> 16) & 1u);
          rowb[k / 2] = (l >> 16) | (h & 0xffff0000u);
        }
        memcpy(&all[(size_t)u * Kb / 2], rowb.data(), Kb / 2 * 4);
      }
      HIP_CHECK(hipMemcpy(dFb, all.data(), (size_t)N * Kb / 2 * 4,
                          hipMemcpyHostToDevice));
    }
    auto kfb_split = [&](int n_hi) {
      launch_kf_mfma_bf16(dFb, dIp, dIdx, dSum, dOrd, dG, dLlh, dLad, dBest,
                          n_hi, Kb, 16, 0.05f, 1e-4f, 0.9999f, 0.f, 1000.f,
                          0);
      launch_kf_bf16(dFb, dIp, dIdx, dSum, dOrd + n_hi, dG, dLlh, dLad,
                     dBest, N - n_hi, Kb, 16, 0.05f, 1e-4f, 0.9999f, 0.f,
                     1000.f, 0);
    };
    for (int thr : thrs) {
      if (only_kfb0 && thr != 0) continue;
      const int n_hi = thr == 0 ? N : (thr == (1 << 30) ? 0 : split_at(thr));
      char name[32];
      snprintf(name, sizeof name, "kfb@%-4d", thr == (1 << 30) ? -1 : thr);
      printf("  (n_mfma=%d)\n", n_hi);
      time3(name, [&] { kfb_split(n_hi); });
    }
    // bf16 separate kernels (phase A of the fused kernel == k1 structure,
    // so k1b approximates the fused kernel's phase-A share)
    if (!only_kfb0) time3("k1b", [&] {
      launch_k1_bf16(dFb, dIp, dIdx, dSum, dOrd, dG, dLlh, N, Kb, 1e-4f,
                     0.9999f, 0);
    });
    if (!only_kfb0) time3("k2b", [&] {
      launch_k2_bf16(dFb, dIp, dIdx, dSum, dG, dLlh, dOrd, dLad, dBest, N,
                     Kb, 16, 0.05f, 1e-4f, 0.9999f, 0.f, 1000.f, 0);
    });
    if (!only_kfb0) time3("k3b", [&] {
      launch_k3_bf16(dFb, dG, dBest, N, Kb, 0.f, 1000.f, 0);
    });
    HIP_CHECK(hipFree(dFb));
  }
  return 0;
}
