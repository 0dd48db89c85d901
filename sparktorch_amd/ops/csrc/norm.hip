// BatchNorm2d / pooling / residual kernels for gfx950 — the ResNet family
// ops (BASELINE config 4: ResNet-18 on synthetic 3x224x224 rows).  The
// reference has no GPU code at all; its conv models run
// eager CPU torch.  Here BN training statistics, normalization (+fused
// ReLU, +fused residual add), BN backward, overlapping max_pool2d, global
// average pool and fused add+relu are hand-written CDNA4 kernels: bf16
// activations, fp32 statistics/params, LDS wave reductions, grids split
// over (channel x batch-slice) so small-C layers still cover all 256 CUs.
//
// Two layout families share the file: the original NCHW kernels (used by
// the generic model converter) and the NHWC (channels-last) kernels the
// fused ResNet path runs on.  NHWC reductions treat x as [M, C] with
// coalesced 8-channel shortx8 groups per thread; BN reductions write block
// partials to scratch and tree-finalize (same-address atomicAdd was
// measured at ~200 ns/round — tools/bn_bench.hip); apply/dx cache
// per-thread scale/shift (6.6 TB/s measured) and recompute the fused-ReLU
// mask from x instead of reading the saved output.

#include "common.h"

// ---------------------------------------------------------------------------
// BatchNorm2d forward statistics.
// Pass 1: grid (C, S) blocks; block (c, s) reduces a slice of the B*HW
// per-channel elements into LDS, then one atomicAdd per block into
// sum[c] / sumsq[c] (fp32, caller-zeroed).
// ---------------------------------------------------------------------------

__global__ void bn_stats_partial_kernel(const bf16raw* __restrict__ x, float* __restrict__ sum,
                                        float* __restrict__ sumsq, int B, int C, int64_t HW) {
  int c = blockIdx.x;
  int64_t total = (int64_t)B * HW;
  int64_t per = ceil_div_i64(total, gridDim.y);
  int64_t lo = (int64_t)blockIdx.y * per;
  int64_t hi = lo + per < total ? lo + per : total;
  float s = 0.f, ss = 0.f;
  for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    int64_t b = i / HW;
    int64_t hw = i - b * HW;
    float v = bf16_to_f32(x[(b * C + c) * HW + hw]);
    s += v;
    ss += v * v;
  }
  __shared__ float ls[256], lss[256];
  ls[threadIdx.x] = s;
  lss[threadIdx.x] = ss;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      ls[threadIdx.x] += ls[threadIdx.x + off];
      lss[threadIdx.x] += lss[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    atomicAdd(&sum[c], ls[0]);
    atomicAdd(&sumsq[c], lss[0]);
  }
}

// Pass 2: mean/invstd for normalization (biased var) + running-stat update
// (unbiased var, torch semantics).  One tiny launch over C.
__global__ void bn_stats_finalize_kernel(const float* __restrict__ sum,
                                         const float* __restrict__ sumsq,
                                         float* __restrict__ mean, float* __restrict__ invstd,
                                         float* __restrict__ running_mean,
                                         float* __restrict__ running_var, int64_t count,
                                         float momentum, float eps, int C) {
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C; c += gridDim.x * blockDim.x) {
    float mu = sum[c] / (float)count;
    float var = sumsq[c] / (float)count - mu * mu;
    var = var > 0.f ? var : 0.f;
    mean[c] = mu;
    invstd[c] = rsqrtf(var + eps);
    if (running_mean != nullptr) {
      float unbiased = count > 1 ? var * (float)count / (float)(count - 1) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
  }
}

extern "C" hipError_t launch_bn_stats(const bf16raw* x, float* sum, float* sumsq, float* mean,
                                      float* invstd, float* running_mean, float* running_var,
                                      int B, int C, int64_t HW, float momentum, float eps,
                                      int nsplit, hipStream_t stream) {
  dim3 grid(C, nsplit);
  bn_stats_partial_kernel<<<grid, 256, 0, stream>>>(x, sum, sumsq, B, C, HW);
  HIP_CHECK_LAUNCH();
  int blocks = (int)ceil_div_i64(C, 256);
  bn_stats_finalize_kernel<<<blocks, 256, 0, stream>>>(sum, sumsq, mean, invstd, running_mean,
                                                       running_var, (int64_t)B * HW, momentum,
                                                       eps, C);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// NHWC BatchNorm statistics: x viewed as [M, C] (M = B*H*W).  Block =
// 64-channel tile x 4 row-groups (coalesced 128-byte row reads); grid
// (ceil(C/64), S) with one atomicAdd per channel per block.
// ---------------------------------------------------------------------------

__global__ void bn_stats_partial_nhwc_kernel(const bf16raw* __restrict__ x,
                                             float* __restrict__ sum, float* __restrict__ sumsq,
                                             int64_t M, int C) {
  int lane_c = threadIdx.x & 63;
  int row_g = threadIdx.x >> 6;  // 0..3
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

// Vectorized stats: thread owns 8 consecutive channels (shortx8 row loads),
// 256/(C/8) row-lanes per block; requires C in {8,16,...,2048} dividing the
// block evenly (ResNet channels are powers of two).
__global__ void bn_stats_partial_nhwc_vec_kernel(const bf16raw* __restrict__ x,
                                                 float* __restrict__ psum,
                                                 float* __restrict__ psumsq, int64_t M, int C) {
  // block s reduces its row slice into partial[s*C + c] with PLAIN stores —
  // atomicAdd contention was measured (tools/bn_bench.hip) to cost ~200 ns
  // per same-address round, dominating the whole kernel at 2048 blocks.
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
      psum[(int64_t)blockIdx.x * C + c0 + j] = a;
      psumsq[(int64_t)blockIdx.x * C + c0 + j] = b;
    }
  }
}

// reduce [S, C] partials and finish mean/invstd + running update.
// One block per channel: 256 threads stride the S partial rows, LDS tree.
__global__ void bn_stats_finalize2_kernel(const float* __restrict__ psum,
                                          const float* __restrict__ psumsq, int S,
                                          float* __restrict__ mean, float* __restrict__ invstd,
                                          float* __restrict__ running_mean,
                                          float* __restrict__ running_var, int64_t count,
                                          float momentum, float eps, int C) {
  int c = blockIdx.x;
  {
    float a = 0.f, b = 0.f;
    for (int sidx = threadIdx.x; sidx < S; sidx += blockDim.x) {
      a += psum[(int64_t)sidx * C + c];
      b += psumsq[(int64_t)sidx * C + c];
    }
    __shared__ float l0[256], l1[256];
    l0[threadIdx.x] = a;
    l1[threadIdx.x] = b;
    __syncthreads();
    for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
      if (threadIdx.x < off) {
        l0[threadIdx.x] += l0[threadIdx.x + off];
        l1[threadIdx.x] += l1[threadIdx.x + off];
      }
      __syncthreads();
    }
    if (threadIdx.x != 0) return;
    float sum = l0[0], sumsq = l1[0];
    float mu = sum / (float)count;
    float var = sumsq / (float)count - mu * mu;
    var = var > 0.f ? var : 0.f;
    mean[c] = mu;
    invstd[c] = rsqrtf(var + eps);
    if (running_mean != nullptr) {
      float unbiased = count > 1 ? var * (float)count / (float)(count - 1) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
  }
}


extern "C" hipError_t launch_bn_stats_nhwc(const bf16raw* x, float* sum, float* sumsq,
                                           float* mean, float* invstd, float* running_mean,
                                           float* running_var, int64_t M, int C, float momentum,
                                           float eps, int nsplit, float* scratch, int S,
                                           hipStream_t stream) {
  if (scratch != nullptr) {
    // vec path: partials in scratch[0 : S*C] (sums) and [S*C : 2*S*C] (sumsq)
    bn_stats_partial_nhwc_vec_kernel<<<S, 256, 0, stream>>>(x, scratch, scratch + (int64_t)S * C,
                                                            M, C);
    HIP_CHECK_LAUNCH();
    bn_stats_finalize2_kernel<<<C, 256, 0, stream>>>(scratch, scratch + (int64_t)S * C, S,
                                                     mean, invstd, running_mean, running_var, M,
                                                     momentum, eps, C);
    HIP_CHECK_LAUNCH();
    return hipSuccess;
  }
  dim3 grid((C + 63) / 64, nsplit);
  bn_stats_partial_nhwc_kernel<<<grid, 256, 0, stream>>>(x, sum, sumsq, M, C);
  HIP_CHECK_LAUNCH();
  int blocks = (int)ceil_div_i64(C, 256);
  bn_stats_finalize_kernel<<<blocks, 256, 0, stream>>>(sum, sumsq, mean, invstd, running_mean,
                                                       running_var, M, momentum, eps, C);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// NHWC apply: elementwise, c = i % C (fully coalesced).
__global__ void bn_apply_nhwc_kernel(const bf16raw* __restrict__ x,
                                     const bf16raw* __restrict__ res, bf16raw* __restrict__ y,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta, int C, int64_t total,
                                     int do_relu) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float v = (bf16_to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (res != nullptr) v += bf16_to_f32(res[i]);
    if (do_relu && v < 0.f) v = 0.f;
    y[i] = f32_to_bf16(v);
  }
}

// 8 elements per thread (C % 8 == 0 keeps the vector inside one channel run)
// Cached-param apply: each thread's channel octet is loop-invariant (the
// grid stride in elements is a multiple of C since C divides 2048), so
// scale/shift fold to 8 registers loaded once — measured 6.6 TB/s vs 1.1
// for per-iteration param loads (tools/bn_bench.hip).
__global__ void bn_apply_nhwc_vec_kernel(const bf16raw* __restrict__ x,
                                         const bf16raw* __restrict__ res,
                                         bf16raw* __restrict__ y, const float* __restrict__ mean,
                                         const float* __restrict__ invstd,
                                         const float* __restrict__ gamma,
                                         const float* __restrict__ beta, int C, int64_t total8,
                                         int do_relu) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int c0 = (int)((i0 << 3) % C);
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float sg = gamma[c0 + j] * invstd[c0 + j];
    sc[j] = sg;
    sh[j] = beta[c0 + j] - mean[c0 + j] * sg;
  }
  for (int64_t i = i0; i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t off = i << 3;
    const shortx8 v = *(const shortx8*)(x + off);
    shortx8 out;
    if (res != nullptr) {
      const shortx8 r8 = *(const shortx8*)(res + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32((bf16raw)v[j]) * sc[j] + sh[j] + bf16_to_f32((bf16raw)r8[j]);
        if (do_relu && f < 0.f) f = 0.f;
        out[j] = (short)f32_to_bf16(f);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32((bf16raw)v[j]) * sc[j] + sh[j];
        if (do_relu && f < 0.f) f = 0.f;
        out[j] = (short)f32_to_bf16(f);
      }
    }
    *(shortx8*)(y + off) = out;
  }
}

extern "C" hipError_t launch_bn_apply_nhwc(const bf16raw* x, const bf16raw* res, bf16raw* y,
                                           const float* mean, const float* invstd,
                                           const float* gamma, const float* beta, int C,
                                           int64_t total, int do_relu, hipStream_t stream) {
  // cached-octet kernel requires the element stride (grid*2048) to be a
  // multiple of C for any grid -> C must divide 2048
  if ((C & 7) == 0 && C <= 2048 && (2048 % C) == 0) {
    int64_t total8 = total >> 3;
    int64_t g = ceil_div_i64(total8, 256);
    int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
    bn_apply_nhwc_vec_kernel<<<grid, 256, 0, stream>>>(x, res, y, mean, invstd, gamma, beta, C,
                                                       total8, do_relu);
  } else {
    int64_t g = ceil_div_i64(total, 256);
    int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
    bn_apply_nhwc_kernel<<<grid, 256, 0, stream>>>(x, res, y, mean, invstd, gamma, beta, C,
                                                   total, do_relu);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// NHWC backward reduce: per-channel Σdy' and Σdy'*xhat (same tile scheme as
// stats; dy' = dy masked by saved post-relu output when fused).
__global__ void bn_bwd_reduce_nhwc_kernel(const bf16raw* __restrict__ dy,
                                          const bf16raw* __restrict__ yrelu,
                                          const bf16raw* __restrict__ x,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ invstd,
                                          float* __restrict__ dbeta, float* __restrict__ dgamma,
                                          int64_t M, int C) {
  int lane_c = threadIdx.x & 63;
  int row_g = threadIdx.x >> 6;
  int c = blockIdx.x * 64 + lane_c;
  int64_t per = ceil_div_i64(M, gridDim.y);
  int64_t lo = (int64_t)blockIdx.y * per;
  int64_t hi = lo + per < M ? lo + per : M;
  float sdy = 0.f, sdyx = 0.f;
  if (c < C) {
    float mu = mean[c], is = invstd[c];
    for (int64_t r = lo + row_g; r < hi; r += 4) {
      int64_t off = r * C + c;
      float g = bf16_to_f32(dy[off]);
      if (yrelu != nullptr && bf16_to_f32(yrelu[off]) <= 0.f) g = 0.f;
      sdy += g;
      sdyx += g * (bf16_to_f32(x[off]) - mu) * is;
    }
  }
  __shared__ float l0[256], l1[256];
  l0[threadIdx.x] = sdy;
  l1[threadIdx.x] = sdyx;
  __syncthreads();
  if (row_g == 0 && c < C) {
    sdy = l0[lane_c] + l0[lane_c + 64] + l0[lane_c + 128] + l0[lane_c + 192];
    sdyx = l1[lane_c] + l1[lane_c + 64] + l1[lane_c + 128] + l1[lane_c + 192];
    atomicAdd(&dbeta[c], sdy);
    atomicAdd(&dgamma[c], sdyx);
  }
}

// relu'd BN backward does NOT read the saved output: the mask is
// recomputed as (gamma*xhat + beta) > 0 from the already-loaded x (2 VALU
// ops instead of a third 2-byte stream).
__global__ void bn_bwd_reduce_nhwc_vec_kernel(const bf16raw* __restrict__ dy, int relu,
                                              const bf16raw* __restrict__ x,
                                              const float* __restrict__ mean,
                                              const float* __restrict__ invstd,
                                              const float* __restrict__ gamma,
                                              const float* __restrict__ beta,
                                              float* __restrict__ pdb,
                                              float* __restrict__ pdg, int64_t M, int C) {
  // partials stored per block (plain stores; see atomic-contention note on
  // the stats kernel).  pdb/pdg are [S, C] scratch.
  int groups = C >> 3;
  int rpg = 256 / groups;
  int cg = threadIdx.x % groups;
  int rg = threadIdx.x / groups;
  int c0 = cg << 3;
  int64_t per = ceil_div_i64(M, gridDim.x);
  int64_t lo = (int64_t)blockIdx.x * per;
  int64_t hi = lo + per < M ? lo + per : M;
  float mu[8], is[8], ga[8], be[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
    ga[j] = gamma[c0 + j];
    be[j] = beta[c0 + j];
  }
  float sdy[8] = {0, 0, 0, 0, 0, 0, 0, 0}, sdyx[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int64_t r = lo + rg; r < hi; r += rpg) {
    int64_t off = r * C + c0;
    const shortx8 g8 = *(const shortx8*)(dy + off);
    const shortx8 x8 = *(const shortx8*)(x + off);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xh = (bf16_to_f32((bf16raw)x8[j]) - mu[j]) * is[j];
      float g = bf16_to_f32((bf16raw)g8[j]);
      if (relu && fmaf(ga[j], xh, be[j]) <= 0.f) g = 0.f;
      sdy[j] += g;
      sdyx[j] += g * xh;
    }
  }
  __shared__ float l0[256][8], l1[256][8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    l0[threadIdx.x][j] = sdy[j];
    l1[threadIdx.x][j] = sdyx[j];
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
      pdb[(int64_t)blockIdx.x * C + c0 + j] = a;
      pdg[(int64_t)blockIdx.x * C + c0 + j] = b;
    }
  }
}

// ACCUMULATES the [S, C] partials into dbeta/dgamma (+=; callers pass
// pre-zeroed buffers — possibly flat-bucket grad views).
__global__ void bn_bwd_reduce_finalize_kernel(const float* __restrict__ pdb,
                                              const float* __restrict__ pdg, int S,
                                              float* __restrict__ dbeta,
                                              float* __restrict__ dgamma, int C) {
  int c = blockIdx.x;
  float a = 0.f, b = 0.f;
  for (int sidx = threadIdx.x; sidx < S; sidx += blockDim.x) {
    a += pdb[(int64_t)sidx * C + c];
    b += pdg[(int64_t)sidx * C + c];
  }
  __shared__ float l0[256], l1[256];
  l0[threadIdx.x] = a;
  l1[threadIdx.x] = b;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      l0[threadIdx.x] += l0[threadIdx.x + off];
      l1[threadIdx.x] += l1[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    dbeta[c] += l0[0];
    dgamma[c] += l1[0];
  }
}

extern "C" hipError_t launch_bn_bwd_reduce_nhwc(const bf16raw* dy, const bf16raw* yrelu,
                                                const bf16raw* x, const float* mean,
                                                const float* invstd, const float* gamma,
                                                const float* beta, float* dbeta,
                                                float* dgamma, int64_t M, int C, int nsplit,
                                                float* scratch, int S, hipStream_t stream) {
  if (scratch != nullptr) {
    bn_bwd_reduce_nhwc_vec_kernel<<<S, 256, 0, stream>>>(dy, yrelu != nullptr ? 1 : 0, x, mean,
                                                         invstd, gamma, beta, scratch,
                                                         scratch + (int64_t)S * C, M, C);
    HIP_CHECK_LAUNCH();
    bn_bwd_reduce_finalize_kernel<<<C, 256, 0, stream>>>(scratch, scratch + (int64_t)S * C, S,
                                                         dbeta, dgamma, C);
    HIP_CHECK_LAUNCH();
    return hipSuccess;
  }
  dim3 grid((C + 63) / 64, nsplit);
  bn_bwd_reduce_nhwc_kernel<<<grid, 256, 0, stream>>>(dy, yrelu, x, mean, invstd, dbeta, dgamma,
                                                      M, C);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

__global__ void bn_bwd_dx_nhwc_kernel(const bf16raw* __restrict__ dy,
                                      const bf16raw* __restrict__ yrelu,
                                      const bf16raw* __restrict__ x,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ dbeta,
                                      const float* __restrict__ dgamma, bf16raw* __restrict__ dx,
                                      int C, int64_t total, float inv_count, int train_stats) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float g = bf16_to_f32(dy[i]);
    if (yrelu != nullptr && bf16_to_f32(yrelu[i]) <= 0.f) g = 0.f;
    float is = invstd[c];
    float out;
    if (train_stats) {
      float xh = (bf16_to_f32(x[i]) - mean[c]) * is;
      out = gamma[c] * is * (g - dbeta[c] * inv_count - xh * dgamma[c] * inv_count);
    } else {
      out = gamma[c] * is * g;
    }
    dx[i] = f32_to_bf16(out);
  }
}

// Cached-param dx.  Per-channel math folded to FOUR coefficients:
//   a  = gamma*invstd
//   d  = gamma*invstd^2*sum_dyxhat/N
//   b2 = gamma*invstd*sum_dy/N - mean*d        (so dx = a*g - b2 - x*d)
//   t  = beta - a*mean                          (relu mask: a*x + t > 0)
// One 256-thread block computes them cooperatively into LDS once (C <= 512
// on this path; every power-of-two C <= 512 divides 2048 so the grid-stride
// octet stays loop-invariant), then every thread reads its fixed octet from
// LDS — the old per-thread 56-scalar-global-load preamble dominated the
// short layers' runtime (l4-class calls sat ~12x off the 3-stream roofline).
__global__ void bn_bwd_dx_nhwc_vec_kernel(const bf16raw* __restrict__ dy, int relu,
                                          const bf16raw* __restrict__ x,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ invstd,
                                          const float* __restrict__ gamma,
                                          const float* __restrict__ beta,
                                          const float* __restrict__ dbeta,
                                          const float* __restrict__ dgamma,
                                          bf16raw* __restrict__ dx, int C, int64_t total8,
                                          float inv_count, int train_stats) {
  __shared__ float sa[512], sb2[512], sd[512], st[512];
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float is = invstd[c];
    float ga = gamma[c];
    float m = mean[c];
    float aa = ga * is;
    float dd = 0.f, bb = 0.f;
    if (train_stats) {
      dd = aa * is * dgamma[c] * inv_count;
      bb = aa * dbeta[c] * inv_count - m * dd;
    }
    sa[c] = aa;
    sd[c] = dd;
    sb2[c] = bb;
    st[c] = (beta != nullptr ? beta[c] : 0.f) - aa * m;
  }
  __syncthreads();

  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int c0 = (int)((i0 << 3) & (C - 1));  // C is a power of two on this path
  float a[8], b2[8], d[8], t[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = sa[c0 + j];
    b2[j] = sb2[c0 + j];
    d[j] = sd[c0 + j];
    t[j] = st[c0 + j];
  }
  for (int64_t i = i0; i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t off = i << 3;
    const shortx8 g8 = *(const shortx8*)(dy + off);
    shortx8 out;
    if (train_stats || relu) {
      const shortx8 x8 = *(const shortx8*)(x + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xv = bf16_to_f32((bf16raw)x8[j]);
        float g = bf16_to_f32((bf16raw)g8[j]);
        if (relu && fmaf(a[j], xv, t[j]) <= 0.f) g = 0.f;
        float o = fmaf(a[j], g, -fmaf(d[j], xv, b2[j]));
        out[j] = (short)f32_to_bf16(o);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        out[j] = (short)f32_to_bf16(a[j] * bf16_to_f32((bf16raw)g8[j]));
    }
    *(shortx8*)(dx + off) = out;
  }
}

extern "C" hipError_t launch_bn_bwd_dx_nhwc(const bf16raw* dy, const bf16raw* yrelu,
                                            const bf16raw* x, const float* mean,
                                            const float* invstd, const float* gamma,
                                            const float* beta, const float* dbeta,
                                            const float* dgamma, bf16raw* dx, int C,
                                            int64_t total, float inv_count, int train_stats,
                                            hipStream_t stream) {
  if ((C & 7) == 0 && C <= 512 && (C & (C - 1)) == 0) {
    int64_t total8 = total >> 3;
    int64_t g = ceil_div_i64(total8, 256);
    int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
    bn_bwd_dx_nhwc_vec_kernel<<<grid, 256, 0, stream>>>(dy, yrelu != nullptr ? 1 : 0, x, mean,
                                                        invstd, gamma, beta, dbeta, dgamma, dx,
                                                        C, total8, inv_count, train_stats);
  } else {
    int64_t g = ceil_div_i64(total, 256);
    int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
    bn_bwd_dx_nhwc_kernel<<<grid, 256, 0, stream>>>(dy, yrelu, x, mean, invstd, gamma, dbeta,
                                                    dgamma, dx, C, total, inv_count,
                                                    train_stats);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// BN apply: y = gamma*(x-mean)*invstd + beta [+ residual] [relu].
// Elementwise over B*C*HW; residual may be null.
// ---------------------------------------------------------------------------

__global__ void bn_apply_kernel(const bf16raw* __restrict__ x, const bf16raw* __restrict__ res,
                                bf16raw* __restrict__ y, const float* __restrict__ mean,
                                const float* __restrict__ invstd, const float* __restrict__ gamma,
                                const float* __restrict__ beta, int C, int64_t HW, int64_t total,
                                int do_relu) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)((i / HW) % C);
    float v = (bf16_to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (res != nullptr) v += bf16_to_f32(res[i]);
    if (do_relu && v < 0.f) v = 0.f;
    y[i] = f32_to_bf16(v);
  }
}

extern "C" hipError_t launch_bn_apply(const bf16raw* x, const bf16raw* res, bf16raw* y,
                                      const float* mean, const float* invstd, const float* gamma,
                                      const float* beta, int C, int64_t HW, int64_t total,
                                      int do_relu, hipStream_t stream) {
  int64_t g = ceil_div_i64(total, 256);
  int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
  bn_apply_kernel<<<grid, 256, 0, stream>>>(x, res, y, mean, invstd, gamma, beta, C, HW, total,
                                            do_relu);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// BN backward reduce: per channel sum of dy' and dy'*xhat, where
// dy' = dy * (y>0) when the forward fused a ReLU (y = saved post-relu out).
// Accumulates straight into dbeta[c] (=Σdy') and dgamma[c] (=Σdy'·xhat) —
// callers pass pre-zeroed fp32 buffers (possibly flat-bucket grad views).
// ---------------------------------------------------------------------------

__global__ void bn_bwd_reduce_kernel(const bf16raw* __restrict__ dy,
                                     const bf16raw* __restrict__ yrelu,
                                     const bf16raw* __restrict__ x, const float* __restrict__ mean,
                                     const float* __restrict__ invstd, float* __restrict__ dbeta,
                                     float* __restrict__ dgamma, int B, int C, int64_t HW) {
  int c = blockIdx.x;
  int64_t total = (int64_t)B * HW;
  int64_t per = ceil_div_i64(total, gridDim.y);
  int64_t lo = (int64_t)blockIdx.y * per;
  int64_t hi = lo + per < total ? lo + per : total;
  float mu = mean[c], is = invstd[c];
  float sdy = 0.f, sdyx = 0.f;
  for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    int64_t b = i / HW;
    int64_t hw = i - b * HW;
    int64_t off = (b * C + c) * HW + hw;
    float g = bf16_to_f32(dy[off]);
    if (yrelu != nullptr && bf16_to_f32(yrelu[off]) <= 0.f) g = 0.f;
    float xh = (bf16_to_f32(x[off]) - mu) * is;
    sdy += g;
    sdyx += g * xh;
  }
  __shared__ float l0[256], l1[256];
  l0[threadIdx.x] = sdy;
  l1[threadIdx.x] = sdyx;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      l0[threadIdx.x] += l0[threadIdx.x + off];
      l1[threadIdx.x] += l1[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    atomicAdd(&dbeta[c], l0[0]);
    atomicAdd(&dgamma[c], l1[0]);
  }
}

extern "C" hipError_t launch_bn_bwd_reduce(const bf16raw* dy, const bf16raw* yrelu,
                                           const bf16raw* x, const float* mean,
                                           const float* invstd, float* dbeta, float* dgamma,
                                           int B, int C, int64_t HW, int nsplit,
                                           hipStream_t stream) {
  dim3 grid(C, nsplit);
  bn_bwd_reduce_kernel<<<grid, 256, 0, stream>>>(dy, yrelu, x, mean, invstd, dbeta, dgamma, B, C,
                                                 HW);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// dx = gamma*invstd * (dy' - Σdy'/N - xhat * Σ(dy'·xhat)/N)   (training)
//    = gamma*invstd * dy'                                      (eval stats)
__global__ void bn_bwd_dx_kernel(const bf16raw* __restrict__ dy, const bf16raw* __restrict__ yrelu,
                                 const bf16raw* __restrict__ x, const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma, const float* __restrict__ dbeta,
                                 const float* __restrict__ dgamma, bf16raw* __restrict__ dx,
                                 int C, int64_t HW, int64_t total, float inv_count,
                                 int train_stats) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)((i / HW) % C);
    float g = bf16_to_f32(dy[i]);
    if (yrelu != nullptr && bf16_to_f32(yrelu[i]) <= 0.f) g = 0.f;
    float is = invstd[c];
    float out;
    if (train_stats) {
      float xh = (bf16_to_f32(x[i]) - mean[c]) * is;
      out = gamma[c] * is * (g - dbeta[c] * inv_count - xh * dgamma[c] * inv_count);
    } else {
      out = gamma[c] * is * g;
    }
    dx[i] = f32_to_bf16(out);
  }
}

extern "C" hipError_t launch_bn_bwd_dx(const bf16raw* dy, const bf16raw* yrelu, const bf16raw* x,
                                       const float* mean, const float* invstd, const float* gamma,
                                       const float* dbeta, const float* dgamma, bf16raw* dx,
                                       int C, int64_t HW, int64_t total, float inv_count,
                                       int train_stats, hipStream_t stream) {
  int64_t g = ceil_div_i64(total, 256);
  int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
  bn_bwd_dx_kernel<<<grid, 256, 0, stream>>>(dy, yrelu, x, mean, invstd, gamma, dbeta, dgamma, dx,
                                             C, HW, total, inv_count, train_stats);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// Fused residual add + relu: out = relu(a + b).  Backward reuses relu_bwd
// (elementwise.hip) against the saved output; both branches receive dz.
// ---------------------------------------------------------------------------

__global__ void add_relu_kernel(const bf16raw* __restrict__ a, const bf16raw* __restrict__ b,
                                bf16raw* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = bf16_to_f32(a[i]) + bf16_to_f32(b[i]);
    out[i] = f32_to_bf16(v > 0.f ? v : 0.f);
  }
}

extern "C" hipError_t launch_add_relu(const bf16raw* a, const bf16raw* b, bf16raw* out, int64_t n,
                                      hipStream_t stream) {
  int64_t g = ceil_div_i64(n, 256);
  int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
  add_relu_kernel<<<grid, 256, 0, stream>>>(a, b, out, n);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// General max_pool2d (overlapping windows, padding) — ResNet stem 3x3/s2/p1.
// argmax stored as uint8 offset (kh*KW+kw) inside the window; backward is a
// gather over the <= ceil(KS/S)^2 windows covering each input element (no
// atomics).
// ---------------------------------------------------------------------------

__global__ void maxpool_gen_fwd_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                                       uint8_t* __restrict__ arg, int64_t BC, int H, int W,
                                       int HO, int WO, int KS, int S, int P) {
  int64_t total = BC * HO * WO;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int wo = (int)(i % WO);
    int ho = (int)((i / WO) % HO);
    int64_t bc = i / ((int64_t)WO * HO);
    const bf16raw* xp = x + bc * H * W;
    float best = -3.4e38f;
    int bestk = 0;
    for (int kh = 0; kh < KS; ++kh) {
      int h = ho * S - P + kh;
      if (h < 0 || h >= H) continue;
      for (int kw = 0; kw < KS; ++kw) {
        int w = wo * S - P + kw;
        if (w < 0 || w >= W) continue;
        float v = bf16_to_f32(xp[h * W + w]);
        if (v > best) {
          best = v;
          bestk = kh * KS + kw;
        }
      }
    }
    y[i] = f32_to_bf16(best);
    arg[i] = (uint8_t)bestk;
  }
}

__global__ void maxpool_gen_bwd_kernel(const bf16raw* __restrict__ dy,
                                       const uint8_t* __restrict__ arg, bf16raw* __restrict__ dx,
                                       int64_t BC, int H, int W, int HO, int WO, int KS, int S,
                                       int P) {
  int64_t total = BC * H * W;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int w = (int)(i % W);
    int h = (int)((i / W) % H);
    int64_t bc = i / ((int64_t)W * H);
    float acc = 0.f;
    // windows (ho,wo) with ho*S - P <= h < ho*S - P + KS
    int ho_lo = (h + P - KS + S) / S;  // ceil((h+P-KS+1)/S) for non-negative
    if (ho_lo < 0) ho_lo = 0;
    int ho_hi = (h + P) / S;
    if (ho_hi >= HO) ho_hi = HO - 1;
    int wo_lo = (w + P - KS + S) / S;
    if (wo_lo < 0) wo_lo = 0;
    int wo_hi = (w + P) / S;
    if (wo_hi >= WO) wo_hi = WO - 1;
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      int kh = h - (ho * S - P);
      if (kh < 0 || kh >= KS) continue;
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        int kw = w - (wo * S - P);
        if (kw < 0 || kw >= KS) continue;
        int64_t o = (bc * HO + ho) * WO + wo;
        if (arg[o] == (uint8_t)(kh * KS + kw)) acc += bf16_to_f32(dy[o]);
      }
    }
    dx[i] = f32_to_bf16(acc);
  }
}

extern "C" hipError_t launch_maxpool_gen_fwd(const bf16raw* x, bf16raw* y, uint8_t* arg,
                                             int64_t BC, int H, int W, int HO, int WO, int KS,
                                             int S, int P, hipStream_t stream) {
  int64_t g = ceil_div_i64(BC * HO * WO, 256);
  int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
  maxpool_gen_fwd_kernel<<<grid, 256, 0, stream>>>(x, y, arg, BC, H, W, HO, WO, KS, S, P);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_maxpool_gen_bwd(const bf16raw* dy, const uint8_t* arg, bf16raw* dx,
                                             int64_t BC, int H, int W, int HO, int WO, int KS,
                                             int S, int P, hipStream_t stream) {
  int64_t g = ceil_div_i64(BC * H * W, 256);
  int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
  maxpool_gen_bwd_kernel<<<grid, 256, 0, stream>>>(dy, arg, dx, BC, H, W, HO, WO, KS, S, P);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// Global average pool: [B,C,H,W] -> [B,C].  One block per (b,c), LDS reduce;
// backward broadcasts dy/HW.
// ---------------------------------------------------------------------------

__global__ void gap_fwd_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                               int64_t HW) {
  int64_t bc = blockIdx.x;
  const bf16raw* xp = x + bc * HW;
  float s = 0.f;
  for (int64_t i = threadIdx.x; i < HW; i += blockDim.x) s += bf16_to_f32(xp[i]);
  __shared__ float ls[256];
  ls[threadIdx.x] = s;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) ls[threadIdx.x] += ls[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) y[bc] = f32_to_bf16(ls[0] / (float)HW);
}

__global__ void gap_bwd_kernel(const bf16raw* __restrict__ dy, bf16raw* __restrict__ dx,
                               int64_t HW, int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    dx[i] = f32_to_bf16(bf16_to_f32(dy[i / HW]) / (float)HW);
  }
}

extern "C" hipError_t launch_gap_fwd(const bf16raw* x, bf16raw* y, int64_t BC, int64_t HW,
                                     hipStream_t stream) {
  gap_fwd_kernel<<<(int)BC, 256, 0, stream>>>(x, y, HW);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_gap_bwd(const bf16raw* dy, bf16raw* dx, int64_t HW, int64_t total,
                                     hipStream_t stream) {
  int64_t g = ceil_div_i64(total, 256);
  int grid = (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
  gap_bwd_kernel<<<grid, 256, 0, stream>>>(dy, dx, HW, total);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}
