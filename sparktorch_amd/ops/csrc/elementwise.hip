// Elementwise / optimizer kernels for gfx950.
//
// Replaces (reference file:line):
//  * fused_adam / fused_sgd — optimizer.step() on CPU per parameter tensor
//    (reference distributed.py:198, server.py:139): ONE grid-stride float4
//    kernel per flat bucket, 1/world grad averaging folded in.
//  * relu_bwd — the dY*(Y>0) mask for the fused linear backward.
//  * bias_grad — column sum of dZ (bf16 -> fp32); small-N variant with
//    8-column thread groups + scratch partials (no atomic chains).
//  * small_wgrad — batched outer product for tiny dW over huge reductions
//    (a 16x32 dW in a 128-row MFMA tile is ~98% dead).
//  * increment_i32 — device step counter for hipGraph-capturable Adam.
//  * cast_f64_f32 — Spark DenseVector rows arrive float64; pack/cast on
//    device (reference does np.stack + .float() on CPU, util.py:87-99).
//
// All memory-bound: vectorized loads (float4 / 8x bf16), grid-stride loops,
// grid capped so the scheduler keeps ~8 blocks/CU (guide §6 G11/G13).

#include "common.h"

static inline int ew_grid(int64_t work_items, int block) {
  int64_t g = ceil_div_i64(work_items, block);
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}

// ---------------------------------------------------------------------------
// fused Adam
// ---------------------------------------------------------------------------

// step_dev != nullptr: read the (already incremented) step from device and
// compute the bias corrections in-kernel — required for hipGraph capture,
// where host-computed scalars would be frozen into the graph.
__global__ void fused_adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                                  float* __restrict__ m, float* __restrict__ v, int64_t n,
                                  float lr, float b1, float b2, float eps, float wd, float bc1,
                                  float bc2, float gscale, int adamw,
                                  const int* __restrict__ step_dev) {
  if (step_dev != nullptr) {
    float t = (float)*step_dev;
    bc1 = 1.f - __powf(b1, t);
    bc2 = 1.f - __powf(b2, t);
  }
  int64_t nvec = n >> 2;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const floatx4* g4 = (const floatx4*)g;
  floatx4* p4 = (floatx4*)p;
  floatx4* m4 = (floatx4*)m;
  floatx4* v4 = (floatx4*)v;
  for (int64_t k = i; k < nvec; k += stride) {
    floatx4 gv = g4[k] * gscale;
    floatx4 pv = p4[k];
    floatx4 mv = m4[k];
    floatx4 vv = v4[k];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gj = gv[j];
      float pj = pv[j];
      if (wd != 0.f) {
        if (adamw)
          pj *= (1.f - lr * wd);
        else
          gj += wd * pj;
      }
      float mj = b1 * mv[j] + (1.f - b1) * gj;
      float vj = b2 * vv[j] + (1.f - b2) * gj * gj;
      float denom = sqrtf(vj / bc2) + eps;
      pj -= (lr / bc1) * (mj / denom);
      pv[j] = pj;
      mv[j] = mj;
      vv[j] = vj;
    }
    p4[k] = pv;
    m4[k] = mv;
    v4[k] = vv;
  }
  // scalar tail
  for (int64_t k = (nvec << 2) + i; k < n; k += stride) {
    float gj = g[k] * gscale;
    float pj = p[k];
    if (wd != 0.f) {
      if (adamw)
        pj *= (1.f - lr * wd);
      else
        gj += wd * pj;
    }
    float mj = b1 * m[k] + (1.f - b1) * gj;
    float vj = b2 * v[k] + (1.f - b2) * gj * gj;
    pj -= (lr / bc1) * (mj / (sqrtf(vj / bc2) + eps));
    p[k] = pj;
    m[k] = mj;
    v[k] = vj;
  }
}

__global__ void increment_i32_kernel(int* x) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *x += 1;
}

extern "C" hipError_t launch_increment_i32(int* x, hipStream_t stream) {
  increment_i32_kernel<<<1, 1, 0, stream>>>(x);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_fused_adam(float* p, const float* g, float* m, float* v, int64_t n,
                                        float lr, float b1, float b2, float eps, float wd,
                                        float bc1, float bc2, float gscale, int adamw,
                                        const int* step_dev, hipStream_t stream) {
  int block = 256;
  fused_adam_kernel<<<ew_grid(n / 4 + 1, block), block, 0, stream>>>(
      p, g, m, v, n, lr, b1, b2, eps, wd, bc1, bc2, gscale, adamw, step_dev);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// fused SGD
// ---------------------------------------------------------------------------

__global__ void fused_sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                                 float* __restrict__ buf, int64_t n, float lr, float momentum,
                                 float wd, float dampening, float gscale, int nesterov, int first,
                                 int has_momentum) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < n; k += stride) {
    float gj = g[k] * gscale;
    float pj = p[k];
    if (wd != 0.f) gj += wd * pj;
    if (has_momentum) {
      float bj = first ? gj : momentum * buf[k] + (1.f - dampening) * gj;
      buf[k] = bj;
      gj = nesterov ? gj + momentum * bj : bj;
    }
    p[k] = pj - lr * gj;
  }
}

extern "C" hipError_t launch_fused_sgd(float* p, const float* g, float* buf, int64_t n, float lr,
                                       float momentum, float wd, float dampening, float gscale,
                                       int nesterov, int first, int has_momentum,
                                       hipStream_t stream) {
  int block = 256;
  fused_sgd_kernel<<<ew_grid(n, block), block, 0, stream>>>(
      p, g, buf, n, lr, momentum, wd, dampening, gscale, nesterov, first, has_momentum);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// ReLU backward mask: dz = dy * (y > 0); bf16 in/out, 8-wide vectorized
// ---------------------------------------------------------------------------

__global__ void relu_bwd_kernel(const bf16raw* __restrict__ dy,
                                const bf16raw* __restrict__ y, bf16raw* __restrict__ dz,
                                int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    dz[i] = bf16_to_f32(y[i]) > 0.f ? dy[i] : (bf16raw)0;
  }
}

__global__ void relu_bwd_vec_kernel(const bf16raw* __restrict__ dy,
                                    const bf16raw* __restrict__ y, bf16raw* __restrict__ dz,
                                    int64_t n8) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t off = i << 3;
    const shortx8 g = *(const shortx8*)(dy + off);
    const shortx8 v = *(const shortx8*)(y + off);
    shortx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      out[j] = bf16_to_f32((bf16raw)v[j]) > 0.f ? g[j] : (short)0;
    *(shortx8*)(dz + off) = out;
  }
}

extern "C" hipError_t launch_relu_bwd(const bf16raw* dy, const bf16raw* y, bf16raw* dz, int64_t n,
                                      hipStream_t stream) {
  int block = 256;
  if ((n & 7) == 0 && n > 0) {
    relu_bwd_vec_kernel<<<ew_grid(n / 8, block), block, 0, stream>>>(dy, y, dz, n >> 3);
  } else {
    relu_bwd_kernel<<<ew_grid(n / 8 + 1, block), block, 0, stream>>>(dy, y, dz, n);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// bias grad: db[n] = sum_m dz[m][n]; dz bf16 row-major [M,N], db fp32 (zeroed)
// grid: (ceil(N/64), m_chunks); each thread owns one column within the chunk
// and strides rows; per-thread partial -> atomicAdd (few chunks => low
// contention).
// ---------------------------------------------------------------------------

// Block = (64, 4): lane x owns 8 consecutive columns (shortx8 vector load),
// consecutive x lanes read consecutive 16 B => fully coalesced row reads; the
// 4 y-lanes stripe rows.  Per-thread fp32 accumulators, LDS-reduce the 4 row
// stripes, one atomicAdd per column per block (guide §6 G12/G13).
__global__ void bias_grad_kernel(const bf16raw* __restrict__ dz, float* __restrict__ db, int M,
                                 int N, int rows_per_chunk) {
  const int n0 = (blockIdx.x * 64 + threadIdx.x) * 8;
  const int m0 = blockIdx.y * rows_per_chunk;
  const int m1 = min(M, m0 + rows_per_chunk);
  float acc[8] = {};
  if (n0 < N) {
    const bool full = (n0 + 8 <= N);
    for (int m = m0 + threadIdx.y; m < m1; m += 4) {
      const bf16raw* row = dz + (int64_t)m * N + n0;
      if (full) {
        shortx8 v = *(const shortx8*)row;
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32((bf16raw)v[j]);
      } else {
        for (int j = 0; j < N - n0; ++j) acc[j] += bf16_to_f32(row[j]);
      }
    }
  }
  // reduce the 4 y-stripes through LDS
  __shared__ float red[4][64][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.y][threadIdx.x][j] = acc[j];
  __syncthreads();
  if (threadIdx.y == 0 && n0 < N) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (n0 + j < N) {
        float s = red[0][threadIdx.x][j] + red[1][threadIdx.x][j] + red[2][threadIdx.x][j] +
                  red[3][threadIdx.x][j];
        atomicAdd(db + n0 + j, s);
      }
    }
  }
}

// Small-N variant (conv bias over huge M): thread owns an 8-column group,
// 256/(N/8) row lanes per block; block partials go to scratch by plain
// stores (no atomic chains), then a tiny tree finalize ACCUMULATES into db.
__global__ void bias_grad_small_kernel(const bf16raw* __restrict__ dz,
                                       float* __restrict__ partial, int64_t M, int N) {
  int groups = N >> 3;
  int rpg = 256 / groups;
  int cg = threadIdx.x % groups;
  int rg = threadIdx.x / groups;
  int c0 = cg << 3;
  int64_t per = ceil_div_i64(M, gridDim.x);
  int64_t lo = (int64_t)blockIdx.x * per;
  int64_t hi = lo + per < M ? lo + per : M;
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int64_t r = lo + rg;
  for (; r + (int64_t)rpg * 8 <= hi; r += (int64_t)rpg * 8) {  // 8 loads in flight
    shortx8 v[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) v[u] = *(const shortx8*)(dz + (r + (int64_t)u * rpg) * N + c0);
#pragma unroll
    for (int u = 0; u < 8; ++u)
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += bf16_to_f32((bf16raw)v[u][j]);
  }
  for (; r < hi; r += rpg) {
    const shortx8 v = *(const shortx8*)(dz + r * N + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += bf16_to_f32((bf16raw)v[j]);
  }
  __shared__ float l0[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) l0[threadIdx.x][j] = s[j];
  __syncthreads();
  if (rg == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float a = 0.f;
      for (int q = 0; q < rpg; ++q) a += l0[q * groups + cg][j];
      partial[(int64_t)blockIdx.x * N + c0 + j] = a;
    }
  }
}

__global__ void bias_grad_small_finalize_kernel(const float* __restrict__ partial, int S,
                                                float* __restrict__ db, int N) {
  int c = blockIdx.x;
  float a = 0.f;
  for (int sidx = threadIdx.x; sidx < S; sidx += blockDim.x) a += partial[(int64_t)sidx * N + c];
  __shared__ float l[256];
  l[threadIdx.x] = a;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) l[threadIdx.x] += l[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) db[c] += l[0];
}

extern "C" hipError_t launch_bias_grad(const bf16raw* dz, float* db, int M, int N,
                                       float* scratch, int S, hipStream_t stream) {
  if (scratch != nullptr) {
    bias_grad_small_kernel<<<S, 256, 0, stream>>>(dz, scratch, M, N);
    HIP_CHECK_LAUNCH();
    bias_grad_small_finalize_kernel<<<N, 256, 0, stream>>>(scratch, S, db, N);
    HIP_CHECK_LAUNCH();
    return hipSuccess;
  }
  dim3 block(64, 4);
  // ~512 row chunks: 2048 waves keeps 256 CUs latency-hidden; per-column
  // atomic chains stay ~512 deep (few us, parallel across columns)
  int rows_per_chunk = (int)ceil_div_i64(M, 512);
  if (rows_per_chunk < 64) rows_per_chunk = 64;
  dim3 grid((unsigned)ceil_div_i64(N, 64 * 8), (unsigned)ceil_div_i64(M, rows_per_chunk));
  bias_grad_kernel<<<grid, block, 0, stream>>>(dz, db, M, N, rows_per_chunk);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// Tiny-output weight gradient: dW[CO,K] = dz^T @ col with CO*K <= 1024 and a
// huge row reduction (e.g. a 16x25 first-layer conv over 37M image rows).
// A 128-row MFMA tile is ~98% dead at M=16, so this is a batched outer
// product instead: stage a row tile of both operands in LDS (contiguous
// copies), every thread accumulates its own output cells over the tile
// (dz reads broadcast within a wave), partials to scratch, tree finalize.
// ---------------------------------------------------------------------------

// One lane owns (co, 8-k-granule); K % 8 == 0 (the conv path pads its col
// matrix), so every c-read is one aligned dwordx4 straight from global (L1
// serves the row re-reads across co lanes) and the z-read broadcasts.  A
// wave covers ceil(CO*K/8/64) cell-groups; remaining waves stripe rows.
__global__ void small_wgrad_kernel(const bf16raw* __restrict__ dz, const bf16raw* __restrict__ col,
                                   float* __restrict__ partial, int64_t M, int CO, int K) {
  const int KG8 = K >> 3;
  const int SLOTS = CO * KG8;               // <= 256
  const int wpc = (SLOTS + 63) >> 6;        // waves covering the cells (1/2/4)
  const int rgs = 4 / wpc;                  // row stripes
  const int wv = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int slot = (wv % wpc) * 64 + lane;
  const int rg = wv / wpc;
  const bool act = slot < SLOTS && rg < rgs;
  const int co = act ? slot / KG8 : 0;
  const int kg = act ? slot % KG8 : 0;

  int64_t per = ceil_div_i64(M, gridDim.x);
  int64_t lo = (int64_t)blockIdx.x * per;
  int64_t hi = lo + per < M ? lo + per : M;

  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (act) {
    const bf16raw* zp = dz + co;
    const bf16raw* cp = col + kg * 8;
    int64_t r = lo + rg;
    // 8-deep row unroll: issue all 16 loads before any FMA so the wave has
    // 8 HBM round-trips in flight (the old 2-deep version was
    // latency-bound at ~1.6 TB/s on the conv1 shape)
    for (; r + rgs * 8 <= hi; r += rgs * 8) {
      float z[8];
      shortx8 c[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        int64_t ru = r + (int64_t)u * rgs;
        z[u] = bf16_to_f32(zp[ru * CO]);
        c[u] = *(const shortx8*)(cp + ru * K);
      }
#pragma unroll
      for (int u = 0; u < 8; ++u)
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) acc[jj] += z[u] * bf16_to_f32((bf16raw)c[u][jj]);
    }
    for (; r < hi; r += rgs) {
      float z = bf16_to_f32(zp[r * CO]);
      const shortx8 c8 = *(const shortx8*)(cp + r * K);
#pragma unroll
      for (int jj = 0; jj < 8; ++jj) acc[jj] += z * bf16_to_f32((bf16raw)c8[jj]);
    }
  }
  // combine the rgs row-stripes of each cell through LDS, then store one
  // partial row per block
  __shared__ float red[256][8];
#pragma unroll
  for (int jj = 0; jj < 8; ++jj) red[threadIdx.x][jj] = acc[jj];
  __syncthreads();
  if (rg == 0 && act) {
#pragma unroll
    for (int jj = 0; jj < 8; ++jj) {
      float a = acc[jj];
      for (int q = 1; q < rgs; ++q) a += red[(wv + q * wpc) * 64 + lane][jj];
      int k = kg * 8 + jj;
      partial[(int64_t)blockIdx.x * CO * K + co * K + k] = a;
    }
  }
}

__global__ void small_wgrad_finalize_kernel(const float* __restrict__ partial, int S,
                                            float* __restrict__ dw, int OUT) {
  int o = blockIdx.x;
  float a = 0.f;
  for (int s = threadIdx.x; s < S; s += blockDim.x) a += partial[(int64_t)s * OUT + o];
  __shared__ float l[256];
  l[threadIdx.x] = a;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) l[threadIdx.x] += l[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) dw[o] += l[0];
}

extern "C" hipError_t launch_small_wgrad(const bf16raw* dz, const bf16raw* col, float* dw,
                                         float* scratch, int S, int64_t M, int CO, int K,
                                         hipStream_t stream) {
  small_wgrad_kernel<<<S, 256, 0, stream>>>(dz, col, scratch, M, CO, K);
  HIP_CHECK_LAUNCH();
  small_wgrad_finalize_kernel<<<CO * K, 256, 0, stream>>>(scratch, S, dw, CO * K);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// fp64 -> fp32 cast (Spark DenseVector pack)
// ---------------------------------------------------------------------------

__global__ void cast_f64_f32_kernel(const double* __restrict__ src, float* __restrict__ dst,
                                    int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < n; k += stride) dst[k] = (float)src[k];
}

extern "C" hipError_t launch_cast_f64_f32(const double* src, float* dst, int64_t n,
                                          hipStream_t stream) {
  int block = 256;
  cast_f64_f32_kernel<<<ew_grid(n, block), block, 0, stream>>>(src, dst, n);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// fp64 -> bf16 in ONE pass (device-side Vector pack: pinned fp64 partition
// staged over PCIe, cast straight to the training dtype — no intermediate
// fp32 tensor, half the HBM traffic of cast_f64_f32 + cast_f32_bf16).
// double2 (16 B) loads keep the read stream on b128 transactions.
typedef double doublex2 __attribute__((ext_vector_type(2)));

__global__ void cast_f64_bf16_kernel(const double* __restrict__ src, bf16raw* __restrict__ dst,
                                     int64_t n) {
  int64_t nvec = n >> 1;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const doublex2* s2 = (const doublex2*)src;
  for (int64_t k = i; k < nvec; k += stride) {
    doublex2 v = s2[k];
    dst[2 * k] = f32_to_bf16((float)v[0]);
    dst[2 * k + 1] = f32_to_bf16((float)v[1]);
  }
  for (int64_t k = (nvec << 1) + i; k < n; k += stride) dst[k] = f32_to_bf16((float)src[k]);
}

extern "C" hipError_t launch_cast_f64_bf16(const double* src, bf16raw* dst, int64_t n,
                                           hipStream_t stream) {
  int block = 256;
  cast_f64_bf16_kernel<<<ew_grid(n / 2 + 1, block), block, 0, stream>>>(src, dst, n);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// f32 <-> bf16 casts (activation ingress), vectorized
// ---------------------------------------------------------------------------

__global__ void cast_f32_bf16_kernel(const float* __restrict__ src, bf16raw* __restrict__ dst,
                                     int64_t n) {
  int64_t nvec = n >> 2;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const floatx4* s4 = (const floatx4*)src;
  shortx4* d4 = (shortx4*)dst;
  for (int64_t k = i; k < nvec; k += stride) {
    floatx4 v = s4[k];
    shortx4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (short)f32_to_bf16(v[j]);
    d4[k] = o;
  }
  for (int64_t k = (nvec << 2) + i; k < n; k += stride) dst[k] = f32_to_bf16(src[k]);
}

extern "C" hipError_t launch_cast_f32_bf16(const float* src, bf16raw* dst, int64_t n,
                                           hipStream_t stream) {
  int block = 256;
  cast_f32_bf16_kernel<<<ew_grid(n / 4 + 1, block), block, 0, stream>>>(src, dst, n);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}
