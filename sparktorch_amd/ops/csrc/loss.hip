// Fused loss kernels (forward + input-gradient in ONE launch).
//
// The training hot loop always needs loss AND dlogits (reference computes
// criterion(...) then loss.backward(), distributed.py:152-176); fusing them
// removes a full re-read of the logits.  Loss is reduced per-block then
// atomicAdd'ed into a single fp32 scalar (guide §6 G12).

#include "common.h"

// ---------------------------------------------------------------------------
// Cross-entropy over bf16 logits [B,C], int64 targets [B].
//   loss_out[0] += sum_b ( log_sum_exp(logits_b) - logit_b[t_b] ) / B
//   dlogits[b][c] = (softmax(logits_b)[c] - 1{c==t_b}) / B
// One thread per row (C is small in this model family: 2..1000); fp32 math.
// ---------------------------------------------------------------------------

__global__ void ce_fused_kernel(const bf16raw* __restrict__ logits,
                                const int64_t* __restrict__ tgt, float* __restrict__ loss_out,
                                bf16raw* __restrict__ dlogits, int B, int C, float inv_B) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  float my_loss = 0.f;
  if (b < B) {
    const bf16raw* row = logits + b * C;
    bf16raw* drow = dlogits + b * C;
    float mx = -1e30f;
    for (int c = 0; c < C; ++c) mx = fmaxf(mx, bf16_to_f32(row[c]));
    float se = 0.f;
    for (int c = 0; c < C; ++c) se += __expf(bf16_to_f32(row[c]) - mx);
    float lse = __logf(se) + mx;
    int64_t t = tgt[b];
    my_loss = (lse - bf16_to_f32(row[t])) * inv_B;
    float inv_se = 1.f / se;
    for (int c = 0; c < C; ++c) {
      float p = __expf(bf16_to_f32(row[c]) - mx) * inv_se;
      if (c == (int)t) p -= 1.f;
      drow[c] = f32_to_bf16(p * inv_B);
    }
  }
  // block reduce loss
  __shared__ float red[256];
  red[threadIdx.x] = my_loss;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0 && red[0] != 0.f) atomicAdd(loss_out, red[0]);
}

// Wave-per-row variant for wide C (e.g. 1000-class ResNet head) or small B:
// 64 lanes sweep the row with shuffle all-reduce for max / sum-exp, so one
// block covers 4 rows on 4 SIMDs instead of 256 rows on one CU.
__global__ void ce_fused_wave_kernel(const bf16raw* __restrict__ logits,
                                     const int64_t* __restrict__ tgt,
                                     float* __restrict__ loss_out,
                                     bf16raw* __restrict__ dlogits, int B, int C, float inv_B) {
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  int64_t b = (int64_t)blockIdx.x * 4 + wave;
  if (b >= B) return;
  const bf16raw* row = logits + b * C;
  bf16raw* drow = dlogits + b * C;
  float mx = -1e30f;
  for (int c = lane; c < C; c += 64) mx = fmaxf(mx, bf16_to_f32(row[c]));
  for (int off = 32; off; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, 64));
  float se = 0.f;
  for (int c = lane; c < C; c += 64) se += __expf(bf16_to_f32(row[c]) - mx);
  for (int off = 32; off; off >>= 1) se += __shfl_xor(se, off, 64);
  int64_t t = tgt[b];
  float inv_se = 1.f / se;
  for (int c = lane; c < C; c += 64) {
    float p = __expf(bf16_to_f32(row[c]) - mx) * inv_se;
    if (c == (int)t) p -= 1.f;
    drow[c] = f32_to_bf16(p * inv_B);
  }
  if (lane == 0) {
    float my_loss = (__logf(se) + mx - bf16_to_f32(row[t])) * inv_B;
    atomicAdd(loss_out, my_loss);
  }
}

extern "C" hipError_t launch_ce_fused(const bf16raw* logits, const int64_t* tgt, float* loss_out,
                                      bf16raw* dlogits, int B, int C, hipStream_t stream) {
  if (C >= 128 || B <= 2048) {
    int grid = (int)ceil_div_i64(B, 4);
    ce_fused_wave_kernel<<<grid, 256, 0, stream>>>(logits, tgt, loss_out, dlogits, B, C,
                                                   1.0f / B);
  } else {
    int block = 256;
    int grid = (int)ceil_div_i64(B, block);
    ce_fused_kernel<<<grid, block, 0, stream>>>(logits, tgt, loss_out, dlogits, B, C, 1.0f / B);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// MSE over bf16 pred/target (mean reduction):
//   loss += (p-t)^2 / n ; dpred = 2(p-t)/n
// ---------------------------------------------------------------------------

__global__ void mse_fused_kernel(const bf16raw* __restrict__ pred,
                                 const bf16raw* __restrict__ target, float* __restrict__ loss_out,
                                 bf16raw* __restrict__ dpred, int64_t n, float inv_n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float my = 0.f;
  for (int64_t k = i; k < n; k += stride) {
    float d = bf16_to_f32(pred[k]) - bf16_to_f32(target[k]);
    my += d * d * inv_n;
    dpred[k] = f32_to_bf16(2.f * d * inv_n);
  }
  __shared__ float red[256];
  red[threadIdx.x] = my;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0 && red[0] != 0.f) atomicAdd(loss_out, red[0]);
}

extern "C" hipError_t launch_mse_fused(const bf16raw* pred, const bf16raw* target, float* loss_out,
                                       bf16raw* dpred, int64_t n, hipStream_t stream) {
  int block = 256;
  int64_t grid = ceil_div_i64(n, block);
  if (grid > 2048) grid = 2048;
  mse_fused_kernel<<<(int)grid, block, 0, stream>>>(pred, target, loss_out, dpred, n, 1.0f / n);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}
