// Conv2d support kernels for gfx950: im2col / col2im (implicit-GEMM
// lowering — the conv GEMMs themselves run on the MFMA kernel in gemm.hip),
// max_pool2d fwd/bwd, and counter-based dropout / dropout2d (mask recomputed
// from (seed, index) in backward — no mask tensor traffic).
//
// Replaces the reference's eager CPU conv path (examples/cnn_network.py
// models run through torch on CPU; reference has no GPU/native code at all).
// All NCHW, bf16 activations.

#include "common.h"

static inline int cgrid(int64_t n, int block) {
  int64_t g = ceil_div_i64(n, block);
  return (int)(g > 4096 ? 4096 : (g < 1 ? 1 : g));
}

// ---------------------------------------------------------------------------
// im2col: X [B,CI,H,W] -> col [B*HO*WO, CI*KH*KW]  (bf16, row-major)
// ---------------------------------------------------------------------------

__global__ void im2col_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ col,
                              int B, int CI, int H, int W, int KH, int KW, int HO, int WO,
                              int sh, int sw, int ph, int pw) {
  int64_t total = (int64_t)B * HO * WO * CI * KH * KW;
  int64_t K = (int64_t)CI * KH * KW;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t m = idx / K;
    int k = (int)(idx % K);
    int kw = k % KW;
    int kh = (k / KW) % KH;
    int ci = k / (KW * KH);
    int wo = (int)(m % WO);
    int ho = (int)((m / WO) % HO);
    int b = (int)(m / ((int64_t)WO * HO));
    int h = ho * sh - ph + kh;
    int w = wo * sw - pw + kw;
    bf16raw v = 0;
    if (h >= 0 && h < H && w >= 0 && w < W)
      v = x[(((int64_t)b * CI + ci) * H + h) * W + w];
    col[idx] = v;
  }
}

extern "C" hipError_t launch_im2col(const bf16raw* x, bf16raw* col, int B, int CI, int H, int W,
                                    int KH, int KW, int HO, int WO, int sh, int sw, int ph,
                                    int pw, hipStream_t stream) {
  int64_t total = (int64_t)B * HO * WO * CI * KH * KW;
  im2col_kernel<<<cgrid(total, 256), 256, 0, stream>>>(x, col, B, CI, H, W, KH, KW, HO, WO, sh,
                                                       sw, ph, pw);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// col2im (gather form, no atomics): dX[b,ci,h,w] = sum over (kh,kw) of
// dcol[(b,ho,wo), (ci,kh,kw)] where ho*sh = h+ph-kh, wo*sw = w+pw-kw.
// ---------------------------------------------------------------------------

__global__ void col2im_kernel(const bf16raw* __restrict__ dcol, bf16raw* __restrict__ dx, int B,
                              int CI, int H, int W, int KH, int KW, int HO, int WO, int sh,
                              int sw, int ph, int pw) {
  int64_t total = (int64_t)B * CI * H * W;
  int64_t K = (int64_t)CI * KH * KW;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int w = (int)(idx % W);
    int h = (int)((idx / W) % H);
    int ci = (int)((idx / ((int64_t)W * H)) % CI);
    int b = (int)(idx / ((int64_t)W * H * CI));
    float acc = 0.f;
    for (int kh = 0; kh < KH; ++kh) {
      int hh = h + ph - kh;
      if (hh < 0 || hh % sh) continue;
      int ho = hh / sh;
      if (ho >= HO) continue;
      for (int kw = 0; kw < KW; ++kw) {
        int ww = w + pw - kw;
        if (ww < 0 || ww % sw) continue;
        int wo = ww / sw;
        if (wo >= WO) continue;
        int64_t m = ((int64_t)b * HO + ho) * WO + wo;
        int k = (ci * KH + kh) * KW + kw;
        acc += bf16_to_f32(dcol[m * K + k]);
      }
    }
    dx[idx] = f32_to_bf16(acc);
  }
}

extern "C" hipError_t launch_col2im(const bf16raw* dcol, bf16raw* dx, int B, int CI, int H,
                                    int W, int KH, int KW, int HO, int WO, int sh, int sw,
                                    int ph, int pw, hipStream_t stream) {
  int64_t total = (int64_t)B * CI * H * W;
  col2im_kernel<<<cgrid(total, 256), 256, 0, stream>>>(dcol, dx, B, CI, H, W, KH, KW, HO, WO, sh,
                                                       sw, ph, pw);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// max_pool2d, kernel == stride (disjoint windows; reference uses
// F.max_pool2d(x, 2), cnn_network.py:18).  argmax stored as u8 window index.
// ---------------------------------------------------------------------------

__global__ void maxpool_fwd_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                                   uint8_t* __restrict__ arg, int BC, int H, int W, int KS,
                                   int HO, int WO) {
  int64_t total = (int64_t)BC * HO * WO;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int wo = (int)(idx % WO);
    int ho = (int)((idx / WO) % HO);
    int bc = (int)(idx / ((int64_t)WO * HO));
    const bf16raw* xp = x + ((int64_t)bc * H + ho * KS) * W + wo * KS;
    float best = -1e30f;
    int bestk = 0;
    for (int i = 0; i < KS; ++i)
      for (int j = 0; j < KS; ++j) {
        float v = bf16_to_f32(xp[i * W + j]);
        if (v > best) {
          best = v;
          bestk = i * KS + j;
        }
      }
    y[idx] = f32_to_bf16(best);
    arg[idx] = (uint8_t)bestk;
  }
}

__global__ void maxpool_bwd_kernel(const bf16raw* __restrict__ dy, const uint8_t* __restrict__ arg,
                                   bf16raw* __restrict__ dx, int BC, int H, int W, int KS, int HO,
                                   int WO) {
  int64_t total = (int64_t)BC * HO * WO;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int wo = (int)(idx % WO);
    int ho = (int)((idx / WO) % HO);
    int bc = (int)(idx / ((int64_t)WO * HO));
    bf16raw* dxp = dx + ((int64_t)bc * H + ho * KS) * W + wo * KS;
    int bestk = arg[idx];
    for (int i = 0; i < KS; ++i)
      for (int j = 0; j < KS; ++j) dxp[i * W + j] = (i * KS + j == bestk) ? dy[idx] : (bf16raw)0;
  }
}

extern "C" hipError_t launch_maxpool_fwd(const bf16raw* x, bf16raw* y, uint8_t* arg, int BC,
                                         int H, int W, int KS, int HO, int WO,
                                         hipStream_t stream) {
  int64_t total = (int64_t)BC * HO * WO;
  maxpool_fwd_kernel<<<cgrid(total, 256), 256, 0, stream>>>(x, y, arg, BC, H, W, KS, HO, WO);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_maxpool_bwd(const bf16raw* dy, const uint8_t* arg, bf16raw* dx,
                                         int BC, int H, int W, int KS, int HO, int WO,
                                         hipStream_t stream) {
  int64_t total = (int64_t)BC * HO * WO;
  maxpool_bwd_kernel<<<cgrid(total, 256), 256, 0, stream>>>(dy, arg, dx, BC, H, W, KS, HO, WO);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// Counter-based dropout: keep iff hash(seed, unit) >= p * 2^32; scale kept
// values by 1/(1-p).  `unit` is the element index (dropout) or the (b,c)
// channel index (dropout2d — zeroes whole channels, like nn.Dropout2d used
// at examples/cnn_network.py:20).  Backward recomputes the mask from the
// same seed: zero mask-memory traffic.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint32_t hash_u32(uint32_t a, uint32_t seed) {
  uint32_t h = a * 0x9E3779B9u + seed;
  h ^= h >> 16;
  h *= 0x85EBCA6Bu;
  h ^= h >> 13;
  h *= 0xC2B2AE35u;
  h ^= h >> 16;
  return h;
}

// unit id: element-wise (units_div=1), NCHW channel-wise (units_div=HW:
// unit = idx/HW = b*C+c), or NHWC channel-wise (units_div=HWC, cmod=C:
// unit = (idx/HWC)*C + idx%C = b*C+c)
__global__ void dropout_kernel(const bf16raw* __restrict__ in, bf16raw* __restrict__ out,
                               int64_t n, int64_t units_div, int64_t cmod, uint32_t thresh,
                               float scale, uint32_t seed) {
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += (int64_t)gridDim.x * blockDim.x) {
    uint32_t unit;
    if (cmod > 0)
      unit = (uint32_t)((idx / units_div) * cmod + idx % cmod);
    else
      unit = (uint32_t)(units_div > 1 ? idx / units_div : idx);
    bool keep = hash_u32(unit, seed) >= thresh;
    out[idx] = keep ? f32_to_bf16(bf16_to_f32(in[idx]) * scale) : (bf16raw)0;
  }
}

// 32-bit index variant: the 64-bit divide in the unit decode is ~70
// instructions on gfx950 and dominated the kernel at CNN sizes
__global__ void dropout_kernel_u32(const bf16raw* __restrict__ in, bf16raw* __restrict__ out,
                                   uint32_t n, uint32_t units_div, uint32_t cmod,
                                   uint32_t thresh, float scale, uint32_t seed) {
  for (uint32_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += gridDim.x * blockDim.x) {
    uint32_t unit;
    if (cmod > 0)
      unit = (idx / units_div) * cmod + idx % cmod;
    else
      unit = units_div > 1 ? idx / units_div : idx;
    bool keep = hash_u32(unit, seed) >= thresh;
    out[idx] = keep ? f32_to_bf16(bf16_to_f32(in[idx]) * scale) : (bf16raw)0;
  }
}

extern "C" hipError_t launch_dropout(const bf16raw* in, bf16raw* out, int64_t n,
                                     int64_t units_div, int64_t cmod, float p, uint32_t seed,
                                     hipStream_t stream) {
  float scale = 1.0f / (1.0f - p);
  uint32_t thresh = (uint32_t)(p * 4294967296.0);
  if (n < (int64_t)1 << 31) {
    dropout_kernel_u32<<<cgrid(n, 256), 256, 0, stream>>>(in, out, (uint32_t)n,
                                                          (uint32_t)units_div, (uint32_t)cmod,
                                                          thresh, scale, seed);
  } else {
    dropout_kernel<<<cgrid(n, 256), 256, 0, stream>>>(in, out, n, units_div, cmod, thresh,
                                                      scale, seed);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}
