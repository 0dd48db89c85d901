// NHWC (channels-last) conv-family kernels for gfx950.
//
// Motivation (profiles/bench_resnet first capture): the NCHW im2col/col2im
// gather kernels were 54% of the ResNet-18 step — 2-byte strided accesses.
// In NHWC every patch element run is CI-contiguous, so im2col/col2im move
// data in 16-byte shortx8 vectors, the implicit-GEMM output [B*HO*WO, CO]
// IS the NHWC activation (no permute kernels at all), and pooling reduces
// over C-contiguous rows.  Layout: x [B, H, W, C] contiguous.
//
// col matrix layout: col[(b*HO+ho)*WO+wo, (kh*KW+kw)*CI+ci]  (row-major,
// K = KH*KW*CI) — matches w2d = w.permute(0,2,3,1).reshape(CO, K).

#include "common.h"

static inline int cgrid8(int64_t n, int block) {
  int64_t g = ceil_div_i64(n, block);
  return (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
}

// ---------------------------------------------------------------------------
// im2col NHWC, vectorized: CI % 8 == 0.  One thread = 8 channels of one
// (b,ho,wo,kh,kw) patch cell; consecutive threads walk ci then (kw,kh) so
// global reads/writes are contiguous runs of CI*2 bytes.
// ---------------------------------------------------------------------------

__global__ void im2col_nhwc_vec_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ col,
                                       int B, int CI, int H, int W, int KH, int KW, int HO,
                                       int WO, int sh, int sw, int ph, int pw, int Kp) {
  int civ8 = CI >> 3;
  int64_t K = Kp;  // row stride (logical K padded to a multiple of 8)
  int64_t total = (int64_t)B * HO * WO * KH * KW * civ8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int ci8 = (int)(idx % civ8);
    int64_t t = idx / civ8;
    int kw = (int)(t % KW);
    t /= KW;
    int kh = (int)(t % KH);
    t /= KH;
    int wo = (int)(t % WO);
    t /= WO;
    int ho = (int)(t % HO);
    int b = (int)(t / HO);
    int h = ho * sh - ph + kh;
    int w = wo * sw - pw + kw;
    int64_t m = ((int64_t)b * HO + ho) * WO + wo;
    shortx8* dst = (shortx8*)(col + m * K + ((int64_t)(kh * KW + kw) * CI) + (ci8 << 3));
    if (h >= 0 && h < H && w >= 0 && w < W) {
      const shortx8* src =
          (const shortx8*)(x + (((int64_t)b * H + h) * W + w) * CI + (ci8 << 3));
      *dst = *src;
    } else {
      shortx8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      *dst = z;
    }
  }
}

// scalar fallback (stem conv: CI=3).  k = idx % K fastest, so for each kh the
// KW*CI stretch of reads and writes is contiguous.
__global__ void im2col_nhwc_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ col,
                                   int B, int CI, int H, int W, int KH, int KW, int HO, int WO,
                                   int sh, int sw, int ph, int pw, int Kp) {
  int64_t K = (int64_t)CI * KH * KW;  // iteration space (row stride is Kp)
  int64_t total = (int64_t)B * HO * WO * K;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int k = (int)(idx % K);
    int64_t m = idx / K;
    int ci = k % CI;
    int kw = (k / CI) % KW;
    int kh = k / (CI * KW);
    int wo = (int)(m % WO);
    int64_t t = m / WO;
    int ho = (int)(t % HO);
    int b = (int)(t / HO);
    int h = ho * sh - ph + kh;
    int w = wo * sw - pw + kw;
    bf16raw v = 0;
    if (h >= 0 && h < H && w >= 0 && w < W)
      v = x[(((int64_t)b * H + h) * W + w) * CI + ci];
    col[m * Kp + k] = v;
  }
}

// Small-CI patch-row copy (the 3-channel stem conv): one thread = one
// (b,ho,wo,kh) patch row of KW*CI elements — reads and writes are contiguous
// runs (42 B for 7x7/CI=3), decode cost amortized over the whole row.
__global__ void im2col_nhwc_rowcopy_kernel(const bf16raw* __restrict__ x,
                                           bf16raw* __restrict__ col, int B, int CI, int H,
                                           int W, int KH, int KW, int HO, int WO, int sh,
                                           int sw, int ph, int pw, int Kp) {
  int64_t K = Kp;
  int64_t total = (int64_t)B * HO * WO * KH;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int kh = (int)(idx % KH);
    int64_t t = idx / KH;
    int wo = (int)(t % WO);
    t /= WO;
    int ho = (int)(t % HO);
    int b = (int)(t / HO);
    int h = ho * sh - ph + kh;
    int w0 = wo * sw - pw;
    int64_t m = ((int64_t)b * HO + ho) * WO + wo;
    bf16raw* dst = col + m * K + (int64_t)kh * KW * CI;
    if (kh == 0) {  // this (m, kh=0) thread also zeros the row's pad columns
      for (int j = KH * KW * CI; j < Kp; ++j) col[m * K + j] = 0;
    }
    if (h < 0 || h >= H) {
      for (int j = 0; j < KW * CI; ++j) dst[j] = 0;
      continue;
    }
    const bf16raw* src = x + (((int64_t)b * H + h) * W + w0) * CI;
    if (w0 >= 0 && w0 + KW <= W) {
      for (int j = 0; j < KW * CI; ++j) dst[j] = src[j];
    } else {
      for (int kw = 0; kw < KW; ++kw) {
        int w = w0 + kw;
        for (int ci = 0; ci < CI; ++ci)
          dst[kw * CI + ci] = (w >= 0 && w < W) ? src[kw * CI + ci] : (bf16raw)0;
      }
    }
  }
}

// LDS-row-staged variant for small CI (the 3-channel stem): one thread owns
// one full output row — scalar gathers land in an LDS scratch row, the
// global write is Kp/8 contiguous shortx8 stores, and adjacent threads
// (adjacent wo) overlap their stride-2 source windows in L1.
__global__ void im2col_nhwc_rowstage_kernel(const bf16raw* __restrict__ x,
                                            bf16raw* __restrict__ col, int B, int CI, int H,
                                            int W, int KH, int KW, int HO, int WO, int sh,
                                            int sw, int ph, int pw, int Kp) {
  extern __shared__ bf16raw rowbuf[];  // [256][Kp]
  bf16raw* my = rowbuf + threadIdx.x * Kp;
  const int KWCI = KW * CI;
  const int K = KH * KWCI;
  int64_t total_m = (int64_t)B * HO * WO;
  for (int64_t m = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; m < total_m;
       m += (int64_t)gridDim.x * blockDim.x) {
    int wo = (int)(m % WO);
    int64_t t = m / WO;
    int ho = (int)(t % HO);
    int b = (int)(t / HO);
    int w0 = wo * sw - pw;
    for (int kh = 0; kh < KH; ++kh) {
      int h = ho * sh - ph + kh;
      bf16raw* dst = my + kh * KWCI;
      if (h < 0 || h >= H) {
        for (int j = 0; j < KWCI; ++j) dst[j] = 0;
        continue;
      }
      const bf16raw* src = x + (((int64_t)b * H + h) * W + w0) * CI;
      if (w0 >= 0 && w0 + KW <= W) {
        for (int j = 0; j < KWCI; ++j) dst[j] = src[j];
      } else {
        for (int kw = 0; kw < KW; ++kw) {
          int w = w0 + kw;
          for (int ci = 0; ci < CI; ++ci)
            dst[kw * CI + ci] = (w >= 0 && w < W) ? src[kw * CI + ci] : (bf16raw)0;
        }
      }
    }
    for (int j = K; j < Kp; ++j) my[j] = 0;
    // own-thread LDS readback: vector store the assembled row
    shortx8* gout = (shortx8*)(col + m * Kp);
    const shortx8* lin = (const shortx8*)my;
    for (int v = 0; v < (Kp >> 3); ++v) gout[v] = lin[v];
  }
}

extern "C" hipError_t launch_im2col_nhwc(const bf16raw* x, bf16raw* col, int B, int CI, int H,
                                         int W, int KH, int KW, int HO, int WO, int sh, int sw,
                                         int ph, int pw, int Kp, hipStream_t stream) {
  if ((CI & 7) == 0) {
    int64_t total = (int64_t)B * HO * WO * KH * KW * (CI >> 3);
    im2col_nhwc_vec_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(x, col, B, CI, H, W, KH, KW,
                                                                   HO, WO, sh, sw, ph, pw, Kp);
  } else if (Kp <= 160) {  // 256 threads x Kp x 2B LDS <= 80 KB -> 2 blocks/CU
    int64_t total_m = (int64_t)B * HO * WO;
    size_t lds = (size_t)256 * Kp * 2;
    im2col_nhwc_rowstage_kernel<<<cgrid8(total_m, 256), 256, lds, stream>>>(
        x, col, B, CI, H, W, KH, KW, HO, WO, sh, sw, ph, pw, Kp);
  } else {
    int64_t total = (int64_t)B * HO * WO * KH;
    im2col_nhwc_rowcopy_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(x, col, B, CI, H, W, KH,
                                                                       KW, HO, WO, sh, sw, ph,
                                                                       pw, Kp);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// col2im NHWC (gather, no atomics), vectorized over 8 channels: for each
// (b,h,w,ci8) accumulate the <= KH*KW dcol cells that cover it.  All reads
// and the write are CI-contiguous.
// ---------------------------------------------------------------------------

__global__ void col2im_nhwc_vec_kernel(const bf16raw* __restrict__ dcol,
                                       bf16raw* __restrict__ dx, int B, int CI, int H, int W,
                                       int KH, int KW, int HO, int WO, int sh, int sw, int ph,
                                       int pw, int Kp) {
  int civ8 = CI >> 3;
  int64_t K = Kp;
  int64_t total = (int64_t)B * H * W * civ8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int ci8 = (int)(idx % civ8);
    int64_t t = idx / civ8;
    int w = (int)(t % W);
    t /= W;
    int h = (int)(t % H);
    int b = (int)(t / H);
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.f;
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
        const shortx8 v = *(const shortx8*)(dcol + m * K + (int64_t)(kh * KW + kw) * CI +
                                            (ci8 << 3));
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf16_to_f32((bf16raw)v[j]);
      }
    }
    shortx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = (short)f32_to_bf16(acc[j]);
    *(shortx8*)(dx + ((((int64_t)b * H + h) * W + w) * CI) + (ci8 << 3)) = out;
  }
}

__global__ void col2im_nhwc_kernel(const bf16raw* __restrict__ dcol, bf16raw* __restrict__ dx,
                                   int B, int CI, int H, int W, int KH, int KW, int HO, int WO,
                                   int sh, int sw, int ph, int pw, int Kp) {
  int64_t K = Kp;
  int64_t total = (int64_t)B * H * W * CI;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int ci = (int)(idx % CI);
    int64_t t = idx / CI;
    int w = (int)(t % W);
    t /= W;
    int h = (int)(t % H);
    int b = (int)(t / H);
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
        acc += bf16_to_f32(dcol[m * K + (int64_t)(kh * KW + kw) * CI + ci]);
      }
    }
    dx[idx] = f32_to_bf16(acc);
  }
}

extern "C" hipError_t launch_col2im_nhwc(const bf16raw* dcol, bf16raw* dx, int B, int CI, int H,
                                         int W, int KH, int KW, int HO, int WO, int sh, int sw,
                                         int ph, int pw, int Kp, hipStream_t stream) {
  if ((CI & 7) == 0) {
    int64_t total = (int64_t)B * H * W * (CI >> 3);
    col2im_nhwc_vec_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(dcol, dx, B, CI, H, W, KH, KW,
                                                                   HO, WO, sh, sw, ph, pw, Kp);
  } else {
    int64_t total = (int64_t)B * H * W * CI;
    col2im_nhwc_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(dcol, dx, B, CI, H, W, KH, KW, HO,
                                                               WO, sh, sw, ph, pw, Kp);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// max_pool2d NHWC (general stride/pad, overlapping windows).  Output index
// has c fastest -> coalesced window reads; backward gathers over covering
// windows (<= ceil(KS/S)^2), no atomics.
// ---------------------------------------------------------------------------

__global__ void maxpool_nhwc_fwd_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                                        uint8_t* __restrict__ arg, int B, int C, int H, int W,
                                        int HO, int WO, int KS, int S, int P) {
  int64_t total = (int64_t)B * HO * WO * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int wo = (int)(t % WO);
    t /= WO;
    int ho = (int)(t % HO);
    int b = (int)(t / HO);
    const bf16raw* xp = x + (int64_t)b * H * W * C + c;
    float best = -3.4e38f;
    int bestk = 0;
    for (int kh = 0; kh < KS; ++kh) {
      int h = ho * S - P + kh;
      if (h < 0 || h >= H) continue;
      for (int kw = 0; kw < KS; ++kw) {
        int w = wo * S - P + kw;
        if (w < 0 || w >= W) continue;
        float v = bf16_to_f32(xp[((int64_t)h * W + w) * C]);
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

__global__ void maxpool_nhwc_bwd_kernel(const bf16raw* __restrict__ dy,
                                        const uint8_t* __restrict__ arg,
                                        bf16raw* __restrict__ dx, int B, int C, int H, int W,
                                        int HO, int WO, int KS, int S, int P) {
  int64_t total = (int64_t)B * H * W * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t t = i / C;
    int w = (int)(t % W);
    t /= W;
    int h = (int)(t % H);
    int b = (int)(t / H);
    float acc = 0.f;
    int ho_lo = (h + P - KS + S) / S;
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
        int64_t o = (((int64_t)b * HO + ho) * WO + wo) * C + c;
        if (arg[o] == (uint8_t)(kh * KS + kw)) acc += bf16_to_f32(dy[o]);
      }
    }
    dx[i] = f32_to_bf16(acc);
  }
}

// Vectorized variants: one thread = 8 channels of one spatial cell (the
// window geometry is channel-invariant, so bounds math amortizes 8x and all
// loads/stores are shortx8).
__global__ void maxpool_nhwc_fwd_vec_kernel(const bf16raw* __restrict__ x,
                                            bf16raw* __restrict__ y, uint8_t* __restrict__ arg,
                                            int B, int C, int H, int W, int HO, int WO, int KS,
                                            int S, int P) {
  int cv = C >> 3;
  int64_t total = (int64_t)B * HO * WO * cv;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c8 = (int)(i % cv);
    int64_t t = i / cv;
    int wo = (int)(t % WO);
    t /= WO;
    int ho = (int)(t % HO);
    int b = (int)(t / HO);
    const bf16raw* xp = x + (int64_t)b * H * W * C + (c8 << 3);
    float best[8];
    int bk[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      best[j] = -3.4e38f;
      bk[j] = 0;
    }
    for (int kh = 0; kh < KS; ++kh) {
      int h = ho * S - P + kh;
      if (h < 0 || h >= H) continue;
      for (int kw = 0; kw < KS; ++kw) {
        int w = wo * S - P + kw;
        if (w < 0 || w >= W) continue;
        const shortx8 v = *(const shortx8*)(xp + ((int64_t)h * W + w) * C);
        int kk = kh * KS + kw;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_to_f32((bf16raw)v[j]);
          if (f > best[j]) {
            best[j] = f;
            bk[j] = kk;
          }
        }
      }
    }
    int64_t o = (((int64_t)b * HO + ho) * WO + wo) * C + (c8 << 3);
    shortx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      out[j] = (short)f32_to_bf16(best[j]);
      arg[o + j] = (uint8_t)bk[j];
    }
    *(shortx8*)(y + o) = out;
  }
}

__global__ void maxpool_nhwc_bwd_vec_kernel(const bf16raw* __restrict__ dy,
                                            const uint8_t* __restrict__ arg,
                                            bf16raw* __restrict__ dx, int B, int C, int H,
                                            int W, int HO, int WO, int KS, int S, int P) {
  int cv = C >> 3;
  int64_t total = (int64_t)B * H * W * cv;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c8 = (int)(i % cv);
    int64_t t = i / cv;
    int w = (int)(t % W);
    t /= W;
    int h = (int)(t % H);
    int b = (int)(t / H);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int ho_lo = (h + P - KS + S) / S;
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
        int64_t o = (((int64_t)b * HO + ho) * WO + wo) * C + (c8 << 3);
        const shortx8 g8 = *(const shortx8*)(dy + o);
        uint8_t kk = (uint8_t)(kh * KS + kw);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (arg[o + j] == kk) acc[j] += bf16_to_f32((bf16raw)g8[j]);
      }
    }
    shortx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = (short)f32_to_bf16(acc[j]);
    *(shortx8*)(dx + ((((int64_t)b * H + h) * W + w) * C) + (c8 << 3)) = out;
  }
}

extern "C" hipError_t launch_maxpool_nhwc_fwd(const bf16raw* x, bf16raw* y, uint8_t* arg, int B,
                                              int C, int H, int W, int HO, int WO, int KS, int S,
                                              int P, hipStream_t stream) {
  if ((C & 7) == 0) {
    int64_t total = (int64_t)B * HO * WO * (C >> 3);
    maxpool_nhwc_fwd_vec_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(x, y, arg, B, C, H, W,
                                                                        HO, WO, KS, S, P);
  } else {
    int64_t total = (int64_t)B * HO * WO * C;
    maxpool_nhwc_fwd_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(x, y, arg, B, C, H, W, HO,
                                                                    WO, KS, S, P);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_maxpool_nhwc_bwd(const bf16raw* dy, const uint8_t* arg, bf16raw* dx,
                                              int B, int C, int H, int W, int HO, int WO, int KS,
                                              int S, int P, hipStream_t stream) {
  if ((C & 7) == 0) {
    int64_t total = (int64_t)B * H * W * (C >> 3);
    maxpool_nhwc_bwd_vec_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(dy, arg, dx, B, C, H, W,
                                                                        HO, WO, KS, S, P);
  } else {
    int64_t total = (int64_t)B * H * W * C;
    maxpool_nhwc_bwd_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(dy, arg, dx, B, C, H, W, HO,
                                                                    WO, KS, S, P);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// Global average pool NHWC: x [B, HW, C] -> y [B, C].  One block per batch
// row; 256 threads sweep channels (coalesced) accumulating over HW.
// ---------------------------------------------------------------------------

__global__ void gap_nhwc_fwd_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ y,
                                    int C, int64_t HW) {
  int b = blockIdx.x;
  const bf16raw* xp = x + (int64_t)b * HW * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float s = 0.f;
    for (int64_t r = 0; r < HW; ++r) s += bf16_to_f32(xp[r * C + c]);
    y[(int64_t)b * C + c] = f32_to_bf16(s / (float)HW);
  }
}

__global__ void gap_nhwc_bwd_kernel(const bf16raw* __restrict__ dy, bf16raw* __restrict__ dx,
                                    int C, int64_t HW, int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t b = i / (HW * C);
    dx[i] = f32_to_bf16(bf16_to_f32(dy[b * C + c]) / (float)HW);
  }
}

extern "C" hipError_t launch_gap_nhwc_fwd(const bf16raw* x, bf16raw* y, int B, int C, int64_t HW,
                                          hipStream_t stream) {
  gap_nhwc_fwd_kernel<<<B, 256, 0, stream>>>(x, y, C, HW);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_gap_nhwc_bwd(const bf16raw* dy, bf16raw* dx, int B, int C,
                                          int64_t HW, hipStream_t stream) {
  int64_t total = (int64_t)B * HW * C;
  gap_nhwc_bwd_kernel<<<cgrid8(total, 256), 256, 0, stream>>>(dy, dx, C, HW, total);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}
