// A/B microbench: split-K wgrad combine strategy on the production shapes.
//   arm A: fp32 atomicAdd epilogue (round-1 production path)
//   arm B: per-slice fp32 slab stores + separate reduce kernel
//          (guide §5 "splitk-seam": slab reducer beats an fp32-atomicAdd
//          accumulator reducer)
// Shapes: ResNet-18 conv wgrads (l1..l4, stem) and the MNIST-MLP fc wgrads
// (with the fused ones-column bias).  Correctness: all-ones operands make
// dw == R exactly in fp32 (R < 2^24).
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/wgrad_ab.hip -o tools/wgrad_ab

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

__global__ void fill_ones_bf16(bf16raw* p, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t s = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += s) p[i] = (bf16raw)0x3F80;
}

static double bench(void (*fn)(void*), void* ctx, int iters) {
  hipEvent_t a, b;
  CK(hipEventCreate(&a));
  CK(hipEventCreate(&b));
  fn(ctx);
  CK(hipDeviceSynchronize());
  CK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i) fn(ctx);
  CK(hipEventRecord(b));
  CK(hipEventSynchronize(b));
  float ms;
  CK(hipEventElapsedTime(&ms, a, b));
  hipEventDestroy(a);
  hipEventDestroy(b);
  return ms / iters * 1000.0;
}

struct Ctx {
  bf16raw *dz, *col;
  float *dw, *db, *ws;
  int CO, N;  // N = Kcol (+1 when bias)
  int64_t R;
  int sk, ones_row;
};

static void run_atomic(void* p) {
  Ctx* c = (Ctx*)p;
  launch_gemm_bf16(c->dz, c->col, 0, c->dw, nullptr, nullptr, c->CO, c->N, (int)c->R, 1, c->CO,
                   c->N - (c->ones_row >= 0 ? 1 : 0) /*sbk = row stride of col*/, 1, EPI_F32,
                   -c->sk, c->ones_row >= 0 ? c->db : nullptr, c->ones_row, 0);
}

static void run_slab(void* p) {
  Ctx* c = (Ctx*)p;
  launch_wgrad_slab(c->dz, c->col, 0, c->dw, c->ones_row >= 0 ? c->db : nullptr, c->CO, c->N,
                    (int)c->R, 1, c->CO, c->N - (c->ones_row >= 0 ? 1 : 0), 1, c->sk,
                    c->ones_row, c->ws, 0, 0);
}

// arm C: transposed — dW^T[Kcol, CO] = x^T dz; reduce writes dw[CO][Kcol].
// A = x^T (sam=1, sak=Kcol), B = dz (sbk=CO, sbn=1).
static void run_slab_t(void* p) {
  Ctx* c = (Ctx*)p;
  int kcol = c->N;  // no-bias shapes only
  launch_wgrad_slab(c->col, c->dz, 0, c->dw, nullptr, kcol, c->CO, (int)c->R, 1, kcol, c->CO, 1,
                    c->sk, -1, c->ws, 1, 0);
}

int main() {
  struct S {
    int CO, Kcol;
    int64_t R;
    int bias;
    const char* name;
  } shapes[] = {
      {64, 576, 802816, 0, "l1 wgrad  "},
      {128, 576, 200704, 0, "l2 wgrad  "},
      {256, 1152, 50176, 0, "l3 wgrad  "},
      {512, 2304, 12544, 0, "l4 wgrad  "},
      {64, 147, 3211264, 0, "stem wgrad"},
      {256, 784, 2097152, 1, "fc1 wgrad "},
      {256, 256, 2097152, 1, "fc2 wgrad "},
  };
  const int sweep[] = {8, 16, 64, 128, 256, 512, 1024};

  for (auto& s : shapes) {
    Ctx c;
    c.CO = s.CO;
    c.R = s.R;
    c.ones_row = s.bias ? s.Kcol : -1;
    c.N = s.Kcol + (s.bias ? 1 : 0);
    size_t ndz = (size_t)s.R * s.CO, ncol = (size_t)s.R * s.Kcol;
    CK(hipMalloc(&c.dz, ndz * 2));
    CK(hipMalloc(&c.col, ncol * 2));
    CK(hipMalloc(&c.dw, (size_t)s.CO * s.Kcol * 4));
    CK(hipMalloc(&c.db, (size_t)s.CO * 4));
    fill_ones_bf16<<<2048, 256>>>(c.dz, ndz);
    fill_ones_bf16<<<2048, 256>>>(c.col, ncol);
    CK(hipDeviceSynchronize());

    // max workspace over the sweep
    size_t max_zs = 1;
    for (int sk : sweep) {
      int zs = wgrad_slab_slices((int)s.R, sk);
      if ((size_t)zs > max_zs) max_zs = zs;
    }
    CK(hipMalloc(&c.ws, max_zs * (size_t)s.CO * c.N * 4));

    double fl = 2.0 * s.CO * s.Kcol * (double)s.R;
    printf("%s CO=%4d Kcol=%4d R=%8ld%s\n", s.name, s.CO, s.Kcol, (long)s.R,
           s.bias ? " +bias" : "");
    for (int sk : sweep) {
      if ((int64_t)sk * 64 > s.R) continue;
      c.sk = sk;
      // correctness once per arm at this sk: dw must equal R
      CK(hipMemset(c.dw, 0, (size_t)s.CO * s.Kcol * 4));
      CK(hipMemset(c.db, 0, (size_t)s.CO * 4));
      run_atomic(&c);
      CK(hipDeviceSynchronize());
      float va, vb = -1;
      CK(hipMemcpy(&va, c.dw + 1234 % (s.CO * s.Kcol), 4, hipMemcpyDeviceToHost));
      if (s.bias) CK(hipMemcpy(&vb, c.db + s.CO / 2, 4, hipMemcpyDeviceToHost));
      bool ok_a = (va == (float)s.R) && (!s.bias || vb == (float)s.R);

      CK(hipMemset(c.dw, 0, (size_t)s.CO * s.Kcol * 4));
      CK(hipMemset(c.db, 0, (size_t)s.CO * 4));
      run_slab(&c);
      CK(hipDeviceSynchronize());
      CK(hipMemcpy(&va, c.dw + 4321 % (s.CO * s.Kcol), 4, hipMemcpyDeviceToHost));
      if (s.bias) CK(hipMemcpy(&vb, c.db + s.CO / 3, 4, hipMemcpyDeviceToHost));
      bool ok_b = (va == (float)s.R) && (!s.bias || vb == (float)s.R);

      bool ok_c = true;
      double tc = -1;
      if (!s.bias) {
        CK(hipMemset(c.dw, 0, (size_t)s.CO * s.Kcol * 4));
        run_slab_t(&c);
        CK(hipDeviceSynchronize());
        CK(hipMemcpy(&va, c.dw + 2718 % (s.CO * s.Kcol), 4, hipMemcpyDeviceToHost));
        ok_c = (va == (float)s.R);
        tc = bench(run_slab_t, &c, 5);
      }
      double ta = bench(run_atomic, &c, 5);
      double tb = bench(run_slab, &c, 5);
      printf("  sk%-4d zs%-4d atomic %8.1fus (%4.0f TF)%s  slab %8.1fus (%4.0f TF)%s  slabT %8.1fus (%4.0f TF)%s\n",
             sk, wgrad_slab_slices((int)s.R, sk), ta, fl / ta / 1e6, ok_a ? "" : " WRONG", tb,
             fl / tb / 1e6, ok_b ? "" : " WRONG", tc, tc > 0 ? fl / tc / 1e6 : 0,
             ok_c ? "" : " WRONG");
    }
    hipFree(c.dz);
    hipFree(c.col);
    hipFree(c.dw);
    hipFree(c.db);
    hipFree(c.ws);
  }
  return 0;
}
