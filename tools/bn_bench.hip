// Standalone microbenchmark for the NHWC BatchNorm kernel variants.
// Build:  hipcc --offload-arch=gfx950 -O3 tools/bn_bench.hip -o gpurun_out/bn_bench
// Run:    ./gpurun_out/bn_bench
// Times each variant on the ResNet-18 layer shapes and prints GB/s so kernel
// changes are chosen from measurement, not guesses.

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../sparktorch_amd/ops/csrc/common.h"

#define CK(x)                                                                  \
  do {                                                                         \
    hipError_t e = (x);                                                        \
    if (e != hipSuccess) {                                                     \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);          \
      exit(1);                                                                 \
    }                                                                          \
  } while (0)

// --------------------------- variant A: scalar column tile ------------------
// (the original bn_stats_partial_nhwc_kernel: 64-channel tile x 4 row lanes)
__global__ void stats_scalar(const bf16raw* __restrict__ x, float* __restrict__ sum,
                             float* __restrict__ sumsq, int64_t M, int C) {
  int lane_c = threadIdx.x & 63;
  int row_g = threadIdx.x >> 6;
  int c = blockIdx.x * 64 + lane_c;
  int64_t per = ceil_div_i64(M, gridDim.y);
  int64_t lo = (int64_t)blockIdx.y * per;
  int64_t hi = lo + per < M ? lo + per : M;
  float s = 0.f, ss = 0.f;
  if (c < C) {
    for (int64_t r = lo + row_g; r < hi; r += 4) {
      float v = bf16_to_f32(x[r * C + c]);
      s += v;
      ss += v * v;
    }
  }
  __shared__ float ls[256], lss[256];
  ls[threadIdx.x] = s;
  lss[threadIdx.x] = ss;
  __syncthreads();
  if (row_g == 0 && c < C) {
    s = ls[lane_c] + ls[lane_c + 64] + ls[lane_c + 128] + ls[lane_c + 192];
    ss = lss[lane_c] + lss[lane_c + 64] + lss[lane_c + 128] + lss[lane_c + 192];
    atomicAdd(&sum[c], s);
    atomicAdd(&sumsq[c], ss);
  }
}

// --------------------------- variant B: current vec (8ch/thread) ------------
__global__ void stats_vec(const bf16raw* __restrict__ x, float* __restrict__ sum,
                          float* __restrict__ sumsq, int64_t M, int C) {
  int groups = C >> 3;
  int rpg = 256 / groups;
  int cg = threadIdx.x % groups;
  int rg = threadIdx.x / groups;
  int c0 = cg << 3;
  int64_t per = ceil_div_i64(M, gridDim.x);
  int64_t lo = (int64_t)blockIdx.x * per;
  int64_t hi = lo + per < M ? lo + per : M;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0}, ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int64_t r = lo + rg; r < hi; r += rpg) {
    const shortx8 v = *(const shortx8*)(x + r * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32((bf16raw)v[j]);
      s[j] += f;
      ss[j] += f * f;
    }
  }
  __shared__ float l0[256][8], l1[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    l0[threadIdx.x][j] = s[j];
    l1[threadIdx.x][j] = ss[j];
  }
  __syncthreads();
  if (rg == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float a = 0.f, b = 0.f;
      for (int q = 0; q < rpg; ++q) {
        a += l0[q * groups + cg][j];
        b += l1[q * groups + cg][j];
      }
      atomicAdd(&sum[c0 + j], a);
      atomicAdd(&sumsq[c0 + j], b);
    }
  }
}

// ------------------- variant C: vec, flat contiguous walk -------------------
// thread t of block b reads flat vector idx = (b*256+t) + iter*grid*256; the
// channel octet follows from the flat index.  Wave reads 1 KB contiguous.
// Accumulate per-channel via LDS at the end (channel octet varies per iter
// only when C > 2048; for C <= 2048 and 256 threads the octet is fixed when
// (256*8) % C == 0 -> i.e. C divides 2048).
__global__ void stats_flat(const bf16raw* __restrict__ x, float* __restrict__ sum,
                           float* __restrict__ sumsq, int64_t total8, int C) {
  // octet index of this thread is fixed across iterations because the
  // per-iteration stride gridDim*256*8 elements is a multiple of C
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0}, ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (int64_t i = i0; i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    const shortx8 v = *(const shortx8*)(x + (i << 3));
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32((bf16raw)v[j]);
      s[j] += f;
      ss[j] += f * f;
    }
  }
  int c0 = (int)((i0 << 3) % C);
  __shared__ float l0[256][8], l1[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    l0[threadIdx.x][j] = s[j];
    l1[threadIdx.x][j] = ss[j];
  }
  __syncthreads();
  // one thread per channel-octet of the block accumulates its peers
  int groups = C >> 3;  // octets per row
  if (threadIdx.x < (unsigned)min(groups, 256)) {
    int cg = threadIdx.x;
    float a[8] = {0, 0, 0, 0, 0, 0, 0, 0}, b[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int t = cg; t < 256; t += groups) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a[j] += l0[t][j];
        b[j] += l1[t][j];
      }
    }
    int cc0 = (int)(((int64_t)blockIdx.x * blockDim.x + cg) << 3) % C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      atomicAdd(&sum[cc0 + j], a[j]);
      atomicAdd(&sumsq[cc0 + j], b[j]);
    }
  }
}

// --------------------------- apply variants ---------------------------------
__global__ void apply_scalar(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                             const float* __restrict__ mean, const float* __restrict__ invstd,
                             const float* __restrict__ gamma, const float* __restrict__ beta,
                             int C, int64_t total, int do_relu) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float v = (bf16_to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (do_relu && v < 0.f) v = 0.f;
    y[i] = f32_to_bf16(v);
  }
}

__global__ void apply_vec(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                          const float* __restrict__ mean, const float* __restrict__ invstd,
                          const float* __restrict__ gamma, const float* __restrict__ beta,
                          int C, int64_t total8, int do_relu) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total8;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t off = i << 3;
    int c0 = (int)(off % C);
    const shortx8 v = *(const shortx8*)(x + off);
    shortx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int c = c0 + j;
      float f = (bf16_to_f32((bf16raw)v[j]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
      if (do_relu && f < 0.f) f = 0.f;
      out[j] = (short)f32_to_bf16(f);
    }
    *(shortx8*)(y + off) = out;
  }
}

// params preloaded: scale/shift precomputed as s=gamma*invstd, b=beta-mean*s,
// loaded once per thread (fixed octet because stride is a multiple of C)
__global__ void apply_vec_cached(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta, int C, int64_t total8,
                                 int do_relu) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int c0 = (int)((i0 << 3) % C);
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float s = gamma[c0 + j] * invstd[c0 + j];
    sc[j] = s;
    sh[j] = beta[c0 + j] - mean[c0 + j] * s;
  }
  for (int64_t i = i0; i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t off = i << 3;
    const shortx8 v = *(const shortx8*)(x + off);
    shortx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32((bf16raw)v[j]) * sc[j] + sh[j];
      if (do_relu && f < 0.f) f = 0.f;
      out[j] = (short)f32_to_bf16(f);
    }
    *(shortx8*)(y + off) = out;
  }
}

// ---------------------------------------------------------------------------

static double bench(void (*run)(void*), void* arg, int iters) {
  hipEvent_t a, b;
  CK(hipEventCreate(&a));
  CK(hipEventCreate(&b));
  run(arg);  // warm
  CK(hipDeviceSynchronize());
  CK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i) run(arg);
  CK(hipEventRecord(b));
  CK(hipEventSynchronize(b));
  float ms;
  CK(hipEventElapsedTime(&ms, a, b));
  return ms / iters * 1000.0;  // us
}

struct Ctx {
  bf16raw* x;
  bf16raw* y;
  float *sum, *sumsq, *mean, *invstd, *gamma, *beta;
  int64_t M;
  int C;
  int grid;
};

int main() {
  struct Shape {
    int64_t M;
    int C;
  } shapes[] = {{802816, 64}, {200704, 128}, {50176, 256}, {12544, 512}};

  for (auto& sp : shapes) {
    Ctx c;
    c.M = sp.M;
    c.C = sp.C;
    int64_t n = sp.M * sp.C;
    CK(hipMalloc(&c.x, n * 2));
    CK(hipMalloc(&c.y, n * 2));
    CK(hipMalloc(&c.sum, sp.C * 4));
    CK(hipMalloc(&c.sumsq, sp.C * 4));
    CK(hipMalloc(&c.mean, sp.C * 4));
    CK(hipMalloc(&c.invstd, sp.C * 4));
    CK(hipMalloc(&c.gamma, sp.C * 4));
    CK(hipMalloc(&c.beta, sp.C * 4));
    CK(hipMemset(c.x, 0x3c, n * 2));
    CK(hipMemset(c.mean, 0, sp.C * 4));
    CK(hipMemset(c.invstd, 0x3f, sp.C * 4));
    CK(hipMemset(c.gamma, 0x3f, sp.C * 4));
    CK(hipMemset(c.beta, 0, sp.C * 4));
    double bytes_stats = (double)n * 2;
    double bytes_apply = (double)n * 4;

    auto t_ss = bench([](void* p) {
      Ctx* c = (Ctx*)p;
      dim3 g((c->C + 63) / 64, 2048 / ((c->C + 63) / 64) > 256 ? 256 : 2048 / ((c->C + 63) / 64));
      hipLaunchKernelGGL(stats_scalar, g, 256, 0, 0, c->x, c->sum, c->sumsq, c->M, c->C);
    }, &c, 20);
    auto t_sv = bench([](void* p) {
      Ctx* c = (Ctx*)p;
      hipLaunchKernelGGL(stats_vec, 2048, 256, 0, 0, c->x, c->sum, c->sumsq, c->M, c->C);
    }, &c, 20);
    auto t_sf = bench([](void* p) {
      Ctx* c = (Ctx*)p;
      int64_t total8 = c->M * c->C / 8;
      int grid = (int)((total8 + 255) / 256);
      if (grid > 8192) grid = 8192;
      hipLaunchKernelGGL(stats_flat, grid, 256, 0, 0, c->x, c->sum, c->sumsq, total8, c->C);
    }, &c, 20);
    auto t_as = bench([](void* p) {
      Ctx* c = (Ctx*)p;
      int64_t total = c->M * c->C;
      int grid = (int)((total + 255) / 256);
      if (grid > 8192) grid = 8192;
      hipLaunchKernelGGL(apply_scalar, grid, 256, 0, 0, c->x, c->y, c->mean, c->invstd, c->gamma,
                         c->beta, c->C, total, 1);
    }, &c, 20);
    auto t_av = bench([](void* p) {
      Ctx* c = (Ctx*)p;
      int64_t total8 = c->M * c->C / 8;
      int grid = (int)((total8 + 255) / 256);
      if (grid > 8192) grid = 8192;
      hipLaunchKernelGGL(apply_vec, grid, 256, 0, 0, c->x, c->y, c->mean, c->invstd, c->gamma,
                         c->beta, c->C, total8, 1);
    }, &c, 20);
    auto t_ac = bench([](void* p) {
      Ctx* c = (Ctx*)p;
      int64_t total8 = c->M * c->C / 8;
      int grid = (int)((total8 + 255) / 256);
      if (grid > 8192) grid = 8192;
      hipLaunchKernelGGL(apply_vec_cached, grid, 256, 0, 0, c->x, c->y, c->mean, c->invstd,
                         c->gamma, c->beta, c->C, total8, 1);
    }, &c, 20);

    printf("M=%7ld C=%3d | stats us: scalar %7.1f vec %7.1f flat %7.1f (GB/s %5.0f %5.0f %5.0f)\n",
           (long)sp.M, sp.C, t_ss, t_sv, t_sf, bytes_stats / t_ss / 1e3, bytes_stats / t_sv / 1e3,
           bytes_stats / t_sf / 1e3);
    printf("                 | apply us: scalar %7.1f vec %7.1f cached %6.1f (GB/s %5.0f %5.0f %5.0f)\n",
           t_as, t_av, t_ac, bytes_apply / t_as / 1e3, bytes_apply / t_av / 1e3,
           bytes_apply / t_ac / 1e3);
    hipFree(c.x);
    hipFree(c.y);
    hipFree(c.sum);
    hipFree(c.sumsq);
    hipFree(c.mean);
    hipFree(c.invstd);
    hipFree(c.gamma);
    hipFree(c.beta);
  }
  return 0;
}
