// Microbenchmark for the MFMA GEMM variants on the ResNet-18 conv shapes.
// Build:  hipcc --offload-arch=gfx950 -O3 tools/gemm_bench.hip -o tools/gemm_bench
// Calls gemm_kernel<...> template instantiations directly so tile variants
// can be compared head-to-head (the production launcher picks one).

#include "../sparktorch_amd/ops/csrc/gemm.hip"

#include <cstdio>
#include <cstdlib>

#define CK(x)                                                                  \
  do {                                                                         \
    hipError_t e = (x);                                                        \
    if (e != hipSuccess) {                                                     \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);          \
      exit(1);                                                                 \
    }                                                                          \
  } while (0)

template <int WR, int WC, bool AG>
double run_fwd(const bf16raw* A, const bf16raw* B, bf16raw* C, int M, int N, int K, int iters) {
  dim3 grid((unsigned)((M + WR * 64 - 1) / (WR * 64)), (unsigned)((N + WC * 64 - 1) / (WC * 64)),
            1);
  hipEvent_t a, b;
  CK(hipEventCreate(&a));
  CK(hipEventCreate(&b));
  auto launch = [&]() {
    gemm_kernel<false, EPI_BF16, false, WR, WC, AG><<<grid, 256>>>(
        A, B, nullptr, C, nullptr, M, N, K, /*sam*/ K, /*sak*/ 1, /*sbk*/ 1, /*sbn*/ K, K,
        nullptr, -1);
  };
  launch();
  CK(hipDeviceSynchronize());
  CK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i) launch();
  CK(hipEventRecord(b));
  CK(hipEventSynchronize(b));
  float ms;
  CK(hipEventElapsedTime(&ms, a, b));
  return ms / iters * 1000.0;
}

// wgrad: dW[M=CO, N=Kcol] = dz^T x col, reduction over R rows, split-K
double run_wgrad(const bf16raw* dz, const bf16raw* col, float* dw, int CO, int Kcol, int64_t R,
                 int splitk, int iters) {
  int kps = (int)((((R + splitk - 1) / splitk) + BK - 1) / BK) * BK;
  int zs = (int)((R + kps - 1) / kps);
  dim3 grid((unsigned)((CO + 127) / 128), (unsigned)((Kcol + 127) / 128), (unsigned)zs);
  hipEvent_t a, b;
  CK(hipEventCreate(&a));
  CK(hipEventCreate(&b));
  auto launch = [&]() {
    // through the production dispatch (selects the 128x256 wide tile for
    // N >= 256 register-staged shapes)
    launch_gemm_bf16(dz, col, 0, dw, nullptr, nullptr, CO, Kcol, (int)R, 1, CO, Kcol, 1,
                     EPI_F32, splitk, nullptr, -1, 0);
  };
  launch();
  CK(hipDeviceSynchronize());
  CK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i) launch();
  CK(hipEventRecord(b));
  CK(hipEventSynchronize(b));
  float ms;
  CK(hipEventElapsedTime(&ms, a, b));
  return ms / iters * 1000.0;
}

int main() {
  struct FS {
    int M, N, K;
    const char* name;
  } fwd[] = {
      {3211264, 64, 147, "stem fwd   "},
      {802816, 64, 576, "l1conv fwd "},
      {802816, 576, 64, "l1 dgrad   "},
      {200704, 128, 576, "l2conv fwd*"},  // first l2 conv (K=64*9)
      {50176, 256, 1152, "l3conv fwd "},
      {12544, 512, 2304, "l4conv fwd "},
  };
  for (auto& s : fwd) {
    size_t an = (size_t)s.M * s.K, bn = (size_t)s.N * s.K, cn = (size_t)s.M * s.N;
    bf16raw *A, *B, *C;
    CK(hipMalloc(&A, an * 2));
    CK(hipMalloc(&B, bn * 2));
    CK(hipMalloc(&C, cn * 2));
    CK(hipMemset(A, 0x3c, an * 2));
    CK(hipMemset(B, 0x3c, bn * 2));
    bool ag = (s.K % 8) == 0;
    double t22 = ag ? run_fwd<2, 2, true>(A, B, C, s.M, s.N, s.K, 10)
                    : run_fwd<2, 2, false>(A, B, C, s.M, s.N, s.K, 10);
    double t41 = s.N <= 64 ? (ag ? run_fwd<4, 1, true>(A, B, C, s.M, s.N, s.K, 10)
                                 : run_fwd<4, 1, false>(A, B, C, s.M, s.N, s.K, 10))
                           : -1;
    double fl = 2.0 * s.M * s.N * s.K;
    printf("%s M=%8d N=%4d K=%5d | 2x2 %8.1fus (%5.0f TF)  4x1 %8.1fus (%5.0f TF)\n", s.name,
           s.M, s.N, s.K, t22, fl / t22 / 1e6, t41, t41 > 0 ? fl / t41 / 1e6 : 0);
    hipFree(A);
    hipFree(B);
    hipFree(C);
  }

  // wgrad split-K sweep (layer1 + layer4 shapes)
  struct WS {
    int CO, Kcol;
    int64_t R;
    const char* name;
  } wg[] = {{64, 576, 802816, "l1 wgrad"}, {512, 2304, 12544, "l4 wgrad"}, {64, 147, 3211264, "stem wgrad"}};
  for (auto& s : wg) {
    bf16raw *dz, *col;
    float* dw;
    CK(hipMalloc(&dz, (size_t)s.R * s.CO * 2));
    CK(hipMalloc(&col, (size_t)s.R * s.Kcol * 2));
    CK(hipMalloc(&dw, (size_t)s.CO * s.Kcol * 4));
    CK(hipMemset(dz, 0x3c, (size_t)s.R * s.CO * 2));
    CK(hipMemset(col, 0x3c, (size_t)s.R * s.Kcol * 2));
    double fl = 2.0 * s.CO * s.Kcol * (double)s.R;
    printf("%s CO=%d Kcol=%d R=%ld |", s.name, s.CO, s.Kcol, (long)s.R);
    for (int sk : {128, 256, 512, 1024}) {
      CK(hipMemset(dw, 0, (size_t)s.CO * s.Kcol * 4));
      double t = run_wgrad(dz, col, dw, s.CO, s.Kcol, s.R, sk, 5);
      printf("  sk%-3d %7.1fus (%4.0f TF)", sk, t, fl / t / 1e6);
    }
    printf("\n");
    hipFree(dz);
    hipFree(col);
    hipFree(dw);
  }
  return 0;
}
