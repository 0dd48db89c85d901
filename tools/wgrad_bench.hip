// Microbench: small_wgrad outer-product kernel vs shapes, plus variants.
// hipcc --offload-arch=gfx950 -O3 tools/wgrad_bench.hip -o tools/wgrad_bench
#include "../sparktorch_amd/ops/csrc/elementwise.hip"
#include <cstdio>
#define CKB(x) do { hipError_t e=(x); if(e!=hipSuccess){printf("err %s %d\n",hipGetErrorString(e),__LINE__);return 1;} } while(0)

int main() {
  // shapes must satisfy the slot kernel's envelope CO*(K/8) <= 256
  struct S { int64_t M; int CO, K; } shapes[] = {
      {37748736, 16, 32}, {21233664, 32, 64}, {2097152, 16, 128}};
  for (auto& s : shapes) {
    bf16raw *dz, *col; float *dw, *scratch;
    CKB(hipMalloc(&dz, s.M * s.CO * 2));
    CKB(hipMalloc(&col, s.M * s.K * 2));
    CKB(hipMalloc(&dw, s.CO * s.K * 4));
    int S = (int)(s.M / 4096); S = S < 256 ? 256 : (S > 2048 ? 2048 : S);
    CKB(hipMalloc(&scratch, (size_t)S * s.CO * s.K * 4));
    CKB(hipMemset(dz, 0x3c, s.M * s.CO * 2));
    CKB(hipMemset(col, 0x3c, s.M * s.K * 2));
    CKB(hipMemset(dw, 0, s.CO * s.K * 4));
    hipEvent_t a, b; CKB(hipEventCreate(&a)); CKB(hipEventCreate(&b));
    launch_small_wgrad(dz, col, dw, scratch, S, s.M, s.CO, s.K, 0);
    CKB(hipDeviceSynchronize());
    CKB(hipEventRecord(a));
    for (int i = 0; i < 5; ++i) launch_small_wgrad(dz, col, dw, scratch, S, s.M, s.CO, s.K, 0);
    CKB(hipEventRecord(b)); CKB(hipEventSynchronize(b));
    float ms; CKB(hipEventElapsedTime(&ms, a, b));
    double bytes = (double)s.M * (s.CO + s.K) * 2;
    printf("M=%9ld CO=%3d K=%3d S=%4d : %8.1f us  (%5.0f GB/s)\n", (long)s.M, s.CO, s.K, S,
           ms / 5 * 1000, bytes / (ms / 5 * 1000) / 1e3);
    // correctness: all-ones-ish input => dw approx M * v^2
    float host; CKB(hipMemcpy(&host, dw, 4, hipMemcpyDeviceToHost));
    printf("   dw[0]=%g (expect ~%g x6)\n", host, (double)s.M * 0.01171875 * 0.01171875 * 6);
    hipFree(dz); hipFree(col); hipFree(dw); hipFree(scratch);
  }
  return 0;
}
