// Implicit-GEMM NHWC conv for gfx950 (stride 1 and 2) — no col matrix.
//
// The explicit im2col path (conv_nhwc.hip) materializes col [M, KH*KW*CI]
// (KH*KW times the activation bytes), streams it through the GEMM, and
// scatters back through col2im for dgrad.  Here the GEMM's A-operand
// staging computes patch addresses directly into a ZERO-RING-PADDED NHWC
// tensor xP [B][H+2P][W+2P][CI]:
//
//   * every 8-element (16 B) k-granule lies inside one (kh,kw) ci-run
//     (CI % 8 == 0), so clean k-chunks stage by async global_load_lds with
//     the usual row-XOR slot swizzle; the K tail (K % BKT != 0) takes a
//     zero-filling register fallback into the same swizzled image;
//   * the pad ring is materialized once per tensor (pad_nhwc) — no patch
//     address is ever out of bounds, no branches in the hot stage;
//   * stride-1 dgrad IS the forward kernel: dx = conv_s1(pad(dz, KH-1-p),
//     flipW) with flipW[ci, (kh,kw), co] = W[co, (KH-1-kh, KW-1-kw), ci]
//     (flip_w2d) — no dcol, no col2im scatter; stride-2 dgrad is a
//     transposed conv and keeps a col2im-only scatter (no im2col);
//   * wgrad keeps gemm.hip's T14 register pipeline but stages its B tile
//     from x patches (CI % 16 == 0) — the duplicated col stream becomes
//     L2-served re-reads of the 1x tensor; split-K combines by fp32
//     atomics or per-slice slabs + reduce (measured per shape).
//
// Eligibility (enforced by the python dispatch): stride in {1, 2},
// CI % 16 == 0, CO % 8 == 0; wmat column-padded to a multiple of 64.
// Everything else keeps the explicit im2col path.

#include "common.h"

#define IC_LDS_PAD 8

typedef shortx8 cfrag_t;

__device__ __forceinline__ int ic_swz(int row, int col) {
  return col ^ (((row >> 4) & 3) << 3);
}

// Patch geometry: maps (output row, k) -> padded-input address.
struct ConvGeom {
  int OW;    // GEMM-space output width ((Wp - KW)/SW + 1)
  int OHW;   // GEMM-space OH * OW
  int SH, SW;  // stride (fwd/wgrad support s>1; dgrad is s1-only)
  int Hp, Wp;  // padded input spatial
  int CI;
  int KW;
  int KWCI;  // KW * CI
  int Kreal; // KH * KW * CI
  // logical output dims: rows with oh >= OHo or ow >= OWo are dropped at the
  // C-write (fwd) / read as zero dz (wgrad).  Equal to OH/OW normally; the
  // stem's symmetric ring over-pads by one row+col and compacts here instead
  // of a strided slice copy.
  int OHo, OWo;
};

// ---------------------------------------------------------------------------
// helpers: pad ring + weight flip
// ---------------------------------------------------------------------------

// xP[b, h+P, w+P, :] = x[b, h, w, :]; ring = 0.  CI % 8 == 0.
__global__ void pad_nhwc_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ xP, int B,
                                int H, int W, int CI, int P) {
  const int Hp = H + 2 * P, Wp = W + 2 * P;
  int64_t total = (int64_t)B * Hp * Wp * (CI >> 3);
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int c8 = (int)(i % (CI >> 3));
    int64_t t = i / (CI >> 3);
    int w = (int)(t % Wp);
    t /= Wp;
    int h = (int)(t % Hp);
    int b = (int)(t / Hp);
    shortx8* dst = (shortx8*)(xP + (((int64_t)b * Hp + h) * Wp + w) * CI + (c8 << 3));
    if (h >= P && h < H + P && w >= P && w < W + P) {
      *dst = *(const shortx8*)(x + (((int64_t)b * H + (h - P)) * W + (w - P)) * CI + (c8 << 3));
    } else {
      shortx8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      *dst = z;
    }
  }
}

extern "C" hipError_t launch_pad_nhwc(const bf16raw* x, bf16raw* xP, int B, int H, int W, int CI,
                                      int P, hipStream_t stream) {
  int64_t total = (int64_t)B * (H + 2 * P) * (W + 2 * P) * (CI >> 3);
  int64_t g = ceil_div_i64(total, 256);
  if (g > 8192) g = 8192;
  pad_nhwc_kernel<<<dim3((unsigned)g), dim3(256), 0, stream>>>(x, xP, B, H, W, CI, P);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// Fused stem space-to-depth: x [B,H,W,3] -> out [B, H/2+2P, W/2+2P, 16]
// with out[b, bh+P, bw+P, (ph*2+pw)*4 + c] = x[b, 2bh+ph, 2bw+pw, c] (c < 3,
// 4th channel zero) and a zero ring of width P — one pass replaces the
// channel pad + 6D permute copy + pad_nhwc chain.
__global__ void s2d_stem_kernel(const bf16raw* __restrict__ x, bf16raw* __restrict__ out, int B,
                                int H, int W, int P) {
  const int Hb = H >> 1, Wb = W >> 1;
  const int Ho = Hb + 2 * P, Wo = Wb + 2 * P;
  // one thread = one 8-channel granule = one phase row (ph) of a 2x2 block
  int64_t total = (int64_t)B * Ho * Wo * 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int ph = (int)(i & 1);
    int64_t t = i >> 1;
    int wo = (int)(t % Wo);
    t /= Wo;
    int ho = (int)(t % Ho);
    int b = (int)(t / Ho);
    shortx8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    int bh = ho - P, bw = wo - P;
    if (bh >= 0 && bh < Hb && bw >= 0 && bw < Wb) {
      const bf16raw* src = x + (((int64_t)b * H + 2 * bh + ph) * W + 2 * bw) * 3;
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        v[c] = (short)src[c];      // pw = 0
        v[4 + c] = (short)src[3 + c];  // pw = 1
      }
    }
    *(shortx8*)(out + ((((int64_t)b * Ho + ho) * Wo + wo) * 16) + ph * 8) = v;
  }
}

extern "C" hipError_t launch_s2d_stem(const bf16raw* x, bf16raw* out, int B, int H, int W, int P,
                                      hipStream_t stream) {
  int64_t total = (int64_t)B * (H / 2 + 2 * P) * (W / 2 + 2 * P) * 2;
  int64_t g = ceil_div_i64(total, 256);
  if (g > 8192) g = 8192;
  s2d_stem_kernel<<<dim3((unsigned)g), dim3(256), 0, stream>>>(x, out, B, H, W, P);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// Wflip[ci][(kh*KW+kw)*CO + co] = W[co][((KH-1-kh)*KW + (KW-1-kw))*CI + ci]
__global__ void flip_w2d_kernel(const bf16raw* __restrict__ w2d, bf16raw* __restrict__ wf,
                                int CO, int CI, int KHW) {
  int64_t total = (int64_t)CO * KHW * CI;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    int co = (int)(i % CO);  // co fastest -> coalesced writes
    int64_t t = i / CO;
    int kk = (int)(t % KHW);
    int ci = (int)(t / KHW);
    wf[((int64_t)ci * KHW + kk) * CO + co] =
        w2d[(int64_t)co * KHW * CI + (int64_t)(KHW - 1 - kk) * CI + ci];
  }
}

extern "C" hipError_t launch_flip_w2d(const bf16raw* w2d, bf16raw* wf, int CO, int CI, int KHW,
                                      hipStream_t stream) {
  int64_t g = ceil_div_i64((int64_t)CO * KHW * CI, 256);
  if (g > 4096) g = 4096;
  flip_w2d_kernel<<<dim3((unsigned)g), dim3(256), 0, stream>>>(w2d, wf, CO, CI, KHW);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// forward / dgrad kernel: Y[M, N] = patches(xP) @ Wmat^T
//   M = B*OH*OW, K = KH*KW*CI (padded to Kpad % BKT == 0 in wmat), N = CO.
// ---------------------------------------------------------------------------

// clean chunk (kt + BKT <= Kreal): one glds per (row, slot); slot decode of
// (kh, kw, ci) — each 16 B granule is inside one ci-run.
template <int ROWS, int BKT>
__device__ __forceinline__ void stage_patch_glds(const bf16raw* __restrict__ xP,
                                                 bf16raw* __restrict__ lds, int row0, int M,
                                                 const ConvGeom g, int kt) {
  constexpr int SLOTS = BKT / 8;
  constexpr int CH_ROWS = 64 / SLOTS;
  constexpr int NCHUNK = ROWS / CH_ROWS;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int r_in = lane / SLOTS;
  const int slot = lane % SLOTS;
#pragma unroll
  for (int c = wid; c < NCHUNK; c += 4) {
    int lrow = c * CH_ROWS + r_in;
    int row = row0 + lrow;
    if (row >= M) row = M - 1;  // clamp: dead rows read valid garbage, never stored
    int b = row / g.OHW;
    int rem = row - b * g.OHW;
    int oh = rem / g.OW;
    int ow = rem - oh * g.OW;
    int sslot = slot ^ (lrow & (SLOTS - 1));
    int k0 = kt + sslot * 8;
    int kh = k0 / g.KWCI;
    int r2 = k0 - kh * g.KWCI;
    int kw = r2 / g.CI;
    int ci = r2 - kw * g.CI;
    const bf16raw* gp =
        xP + (((int64_t)b * g.Hp + oh * g.SH + kh) * g.Wp + ow * g.SW + kw) * g.CI + ci;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gp,
        (__attribute__((address_space(3))) unsigned int*)(lds + c * CH_ROWS * BKT), 16, 0, 0);
  }
}

// K-tail chunk: per-granule register loads with zero fill past Kreal, same
// swizzled LDS image.
template <int ROWS, int BKT>
__device__ __forceinline__ void stage_patch_tail(const bf16raw* __restrict__ xP,
                                                 bf16raw* __restrict__ lds, int row0, int M,
                                                 const ConvGeom g, int kt) {
  constexpr int SLOTS = BKT / 8;
  constexpr int GRANULES = ROWS * SLOTS;
  constexpr int PER_T = GRANULES / 256;
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < PER_T; ++i) {
    int idx = t * PER_T + i;
    int lrow = idx / SLOTS;
    int slot = idx % SLOTS;
    int row = row0 + lrow;
    if (row >= M) row = M - 1;
    int b = row / g.OHW;
    int rem = row - b * g.OHW;
    int oh = rem / g.OW;
    int ow = rem - oh * g.OW;
    int k0 = kt + slot * 8;
    bf16raw* dst = lds + lrow * BKT + ((slot ^ (lrow & (SLOTS - 1))) << 3);
    if (k0 + 8 <= g.Kreal) {
      int kh = k0 / g.KWCI;
      int r2 = k0 - kh * g.KWCI;
      int kw = r2 / g.CI;
      int ci = r2 - kw * g.CI;
      *(shortx8*)dst = *(const shortx8*)(xP + (((int64_t)b * g.Hp + oh * g.SH + kh) * g.Wp +
                                                ow * g.SW + kw) * g.CI + ci);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int k = k0 + j;
        bf16raw v = 0;
        if (k < g.Kreal) {
          int kh = k / g.KWCI;
          int r2 = k - kh * g.KWCI;
          int kw = r2 / g.CI;
          int ci = r2 - kw * g.CI;
          v = xP[(((int64_t)b * g.Hp + oh * g.SH + kh) * g.Wp + ow * g.SW + kw) * g.CI + ci];
        }
        dst[j] = v;
      }
    }
  }
}

// B: Wmat rows (k-contiguous, column-padded to Kpad) — standard glds with
// the same swizzled image; rows clamp to N-1 (finite garbage, masked at
// the C-write) so N need not be a tile multiple.
template <int ROWS, int BKT>
__device__ __forceinline__ void stage_w_glds(const bf16raw* __restrict__ src,
                                             bf16raw* __restrict__ lds, int row0, int N,
                                             int64_t srow, int kt) {
  constexpr int SLOTS = BKT / 8;
  constexpr int CH_ROWS = 64 / SLOTS;
  constexpr int NCHUNK = ROWS / CH_ROWS;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int r_in = lane / SLOTS;
  const int slot = lane % SLOTS;
#pragma unroll
  for (int c = wid; c < NCHUNK; c += 4) {
    int row = c * CH_ROWS + r_in;
    int grow = row0 + row;
    if (grow >= N) grow = N - 1;
    int sslot = slot ^ (row & (SLOTS - 1));
    const bf16raw* g = src + (int64_t)grow * srow + kt + sslot * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds + c * CH_ROWS * BKT), 16, 0, 0);
  }
}

// NFR = active 16-column output fragments per wave (4 = the full 64-wide
// sub-tile; 1 serves N <= 16 shapes — e.g. a CI=16 dgrad — without
// issuing 75% dead MFMAs).
template <int WR, int WC, bool RELU, bool BIAS, int NFR = 4>
__global__ __launch_bounds__(WR * WC * 64, 2) void conv_implicit_fwd_kernel(
    const bf16raw* __restrict__ xP, const bf16raw* __restrict__ wmat, bf16raw* __restrict__ y,
    const float* __restrict__ bias, int M, int N, int Kpad, ConvGeom g) {
  constexpr int BMt = WR * 64;
  constexpr int BNt = WC * 64;
  constexpr int BKT = (WC == 1) ? 32 : 64;
  constexpr int SUBS = BKT / 32;
  constexpr int SLOTS = BKT / 8;
  __shared__ bf16raw As[2][BMt * BKT];
  __shared__ bf16raw Bs[2][BNt * BKT];

  // XCD-aware bijective block swizzle (same as gemm.hip)
  const int gx = gridDim.x;
  int nwg = gx * gridDim.y;
  int orig = blockIdx.y * gx + blockIdx.x;
  int q = nwg >> 3, rr = nwg & 7;
  int wg = ((orig & 7) < rr ? (orig & 7) * (q + 1) : rr * (q + 1) + ((orig & 7) - rr) * q) +
           (orig >> 3);
  const int m0 = (wg % gx) * BMt;
  const int n0 = (wg / gx) * BNt;

  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wr = wid / WC;
  const int wc = wid % WC;
  const int l15 = lane & 15, kg = lane >> 4;

  floatx4 acc[4][NFR] = {};

#define IC_STAGE(bufi, kt)                                                 \
  do {                                                                     \
    if ((kt) + BKT <= g.Kreal)                                             \
      stage_patch_glds<BMt, BKT>(xP, As[bufi], m0, M, g, kt);              \
    else                                                                   \
      stage_patch_tail<BMt, BKT>(xP, As[bufi], m0, M, g, kt);              \
    stage_w_glds<BNt, BKT>(wmat, Bs[bufi], n0, N, Kpad, kt);               \
  } while (0)

  int buf = 0;
  IC_STAGE(0, 0);
  __syncthreads();

  for (int kt = 0; kt < Kpad; kt += BKT) {
    if (kt + BKT < Kpad) IC_STAGE(buf ^ 1, kt + BKT);

#pragma unroll
    for (int sub = 0; sub < SUBS; ++sub) {
      cfrag_t a[4], b[NFR];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr * 64 + mi * 16 + l15;
        int kq = kg + sub * 4;
        a[mi] = *(const cfrag_t*)&As[buf][row * BKT + ((kq ^ (row & (SLOTS - 1))) << 3)];
      }
#pragma unroll
      for (int ni = 0; ni < NFR; ++ni) {
        int rowb = wc * 64 + ni * 16 + l15;
        int kq = kg + sub * 4;
        b[ni] = *(const cfrag_t*)&Bs[buf][rowb * BKT + ((kq ^ (rowb & (SLOTS - 1))) << 3)];
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < NFR; ++ni)
          acc[mi][ni] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }

    __syncthreads();
    buf ^= 1;
  }
#undef IC_STAGE

  // bf16 epilogue via per-wave LDS transpose (one contiguous 16 B store per
  // lane) — same layout trick as gemm.hip's bf16 epilogue.  N % 8 == 0.
  const int m_base = m0 + wr * 64;
  const int n_base = n0 + wc * 64;
  __syncthreads();
  constexpr int EPAD = 68;
  float* ep = (float*)&As[0][0] + wid * 16 * EPAD;
  const int orow = lane >> 2;
  const int oct = lane & 3;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NFR; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) ep[(kg * 4 + r) * EPAD + ni * 16 + l15] = acc[mi][ni][r];
    __builtin_amdgcn_s_waitcnt(0);
    int m = m_base + mi * 16 + orow;
    int64_t mout = m;
    if (g.OWo != g.OW || g.OHo * g.OW != g.OHW) {
      int b = m / g.OHW;
      int rem = m - b * g.OHW;
      int oh = rem / g.OW;
      int ow = rem - oh * g.OW;
      mout = (oh < g.OHo && ow < g.OWo) ? (((int64_t)b * g.OHo + oh) * g.OWo + ow) : -1;
    }
    if (m < M && mout >= 0) {
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        int c0 = oct * 16 + h * 8;
        int n = n_base + c0;
        if (c0 < NFR * 16 && n < N) {
          alignas(16) short outp[8];
          const float* src = ep + orow * EPAD + c0;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float val = src[j];
            if (BIAS) val += bias[n + j];
            if (RELU) val = fmaxf(val, 0.f);
            outp[j] = (short)f32_to_bf16(val);
          }
          *(shortx8*)(y + mout * N + n) = *(const shortx8*)outp;
        }
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
  }
}

extern "C" hipError_t launch_conv_implicit_fwd(const bf16raw* xP, const bf16raw* wmat,
                                               const float* bias, bf16raw* y, int B, int Hp,
                                               int Wp, int CI, int KH, int KW, int SH, int SW,
                                               int N, int Kpad, int relu, int OHo, int OWo,
                                               hipStream_t stream) {
  const int OH = (Hp - KH) / SH + 1, OW = (Wp - KW) / SW + 1;
  const int M = B * OH * OW;
  ConvGeom g;
  g.OW = OW;
  g.OHW = OH * OW;
  g.Hp = Hp;
  g.Wp = Wp;
  g.CI = CI;
  g.KW = KW;
  g.KWCI = KW * CI;
  g.Kreal = KH * KW * CI;
  g.SH = SH;
  g.SW = SW;
  g.OHo = OHo > 0 ? OHo : OH;
  g.OWo = OWo > 0 ? OWo : OW;
  const bool narrow = (N <= 64);
  const int bm = narrow ? 256 : 128, bn = narrow ? 64 : 128;
  dim3 grid((unsigned)ceil_div_i64(M, bm), (unsigned)ceil_div_i64(N, bn), 1);
  dim3 block(256);

  // active fragment columns: N <= 16 -> 1, N <= 32 -> 2, else the full 64
  // (small-CO shapes otherwise issue 50-75% dead MFMAs in the narrow tile)
#define IC_DISPATCH(RELUV, BIASV)                                                           \
  do {                                                                                      \
    if (narrow && N <= 16)                                                                  \
      conv_implicit_fwd_kernel<4, 1, RELUV, BIASV, 1><<<grid, block, 0, stream>>>(          \
          xP, wmat, y, bias, M, N, Kpad, g);                                                \
    else if (narrow && N <= 32)                                                             \
      conv_implicit_fwd_kernel<4, 1, RELUV, BIASV, 2><<<grid, block, 0, stream>>>(          \
          xP, wmat, y, bias, M, N, Kpad, g);                                                \
    else if (narrow)                                                                        \
      conv_implicit_fwd_kernel<4, 1, RELUV, BIASV><<<grid, block, 0, stream>>>(             \
          xP, wmat, y, bias, M, N, Kpad, g);                                                \
    else                                                                                    \
      conv_implicit_fwd_kernel<2, 2, RELUV, BIASV><<<grid, block, 0, stream>>>(             \
          xP, wmat, y, bias, M, N, Kpad, g);                                                \
  } while (0)

  if (relu && bias) IC_DISPATCH(true, true);
  else if (relu) IC_DISPATCH(true, false);
  else if (bias) IC_DISPATCH(false, true);
  else IC_DISPATCH(false, false);
#undef IC_DISPATCH
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------------------
// wgrad: dW[CO, KH*KW*CI] = dz^T @ patches(xP), split-K over batch rows.
// T14 register pipeline; B rows are patch slices read straight from xP.
// ---------------------------------------------------------------------------

// A slab chunk: [ROWS m-rows][4096/ROWS k] of dz^T starting at column kt
// (dz rows kt.., cols m0..m0+ROWS-1).  256 threads x 16 elements.
template <int ROWS>
__device__ __forceinline__ void wg_stage_a(const bf16raw* __restrict__ dz, int m0, int CO,
                                           int kt, int kmax, const ConvGeom g,
                                           bf16raw* __restrict__ regs, int t) {
  constexpr int TPK = ROWS / 16;  // threads per k-column
  int k = t / TPK;
  int r0 = (t % TPK) * 16;
  int gk = kt + k;
  int64_t gkc = gk;
  if (g.OWo != g.OW || g.OHo * g.OW != g.OHW) {
    int b = gk / g.OHW;
    int rem = gk - b * g.OHW;
    int oh = rem / g.OW;
    int ow = rem - oh * g.OW;
    gkc = (oh < g.OHo && ow < g.OWo) ? (((int64_t)b * g.OHo + oh) * g.OWo + ow) : -1;
  }
  if (gk < kmax && gkc >= 0) {
    const bf16raw* base = dz + gkc * CO;
    int rem = CO - m0 - r0;
    if (rem >= 16) {
      const bf16raw* sp = base + m0 + r0;
      *(shortx8*)regs = *(const shortx8*)sp;
      *(shortx8*)(regs + 8) = *(const shortx8*)(sp + 8);
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j)
        regs[j] = (j < rem) ? base[m0 + r0 + j] : (bf16raw)0;
    }
  } else {
#pragma unroll
    for (int j = 0; j < 16; ++j) regs[j] = 0;
  }
}

// B slab chunk: [ROWS n-cols][4096/ROWS k] — n = (kh,kw,ci), k = batch row.
// A thread's 16-n span stays inside one ci-run (CI % 16 == 0, n0 16-aligned).
template <int ROWS>
__device__ __forceinline__ void wg_stage_b(const bf16raw* __restrict__ xP, int n0, int Nmax,
                                           int kt, int kmax, const ConvGeom g,
                                           bf16raw* __restrict__ regs, int t) {
  constexpr int TPK = ROWS / 16;
  int k = t / TPK;
  int r0 = (t % TPK) * 16;
  int gk = kt + k;  // batch row
  int n = n0 + r0;
  if (gk < kmax && n < Nmax) {
    int b = gk / g.OHW;
    int rem = gk - b * g.OHW;
    int oh = rem / g.OW;
    int ow = rem - oh * g.OW;
    int kh = n / g.KWCI;
    int r2 = n - kh * g.KWCI;
    int kw = r2 / g.CI;
    int ci = r2 - kw * g.CI;
    const bf16raw* sp =
        xP + (((int64_t)b * g.Hp + oh * g.SH + kh) * g.Wp + ow * g.SW + kw) * g.CI + ci;
    *(shortx8*)regs = *(const shortx8*)sp;
    *(shortx8*)(regs + 8) = *(const shortx8*)(sp + 8);
  } else {
#pragma unroll
    for (int j = 0; j < 16; ++j) regs[j] = 0;
  }
}

// Store a staged chunk into the padded swizzled image.  col0 = the chunk's
// absolute column base inside the BKT-wide slab; the XOR swizzle applies
// within each 32-column half (matching the fragment-read addressing).
template <int ROWS, int LDSTRIDE>
__device__ __forceinline__ void wg_stage_write(bf16raw* __restrict__ lds,
                                               const bf16raw* __restrict__ regs, int t,
                                               int col0) {
  constexpr int TPK = ROWS / 16;
  int c = col0 + t / TPK;
  int r0 = (t % TPK) * 16;
  int half = c & ~31;
  int cl = c & 31;
#pragma unroll
  for (int j = 0; j < 16; ++j)
    lds[(r0 + j) * LDSTRIDE + half + ic_swz(r0 + j, cl)] = regs[j];
}

// MFR = active 16-row output fragments per wave (4 = full 64 rows; 2
// serves CO <= 32 without half-dead MFMAs).
template <bool SLAB, int WR, int WC, int MFR = 4>
__global__ __launch_bounds__(256, 2) void conv_implicit_wgrad_kernel(
    const bf16raw* __restrict__ dz, const bf16raw* __restrict__ xP, float* __restrict__ out,
    int CO, int N, int K, ConvGeom g, int k_per_split) {
  constexpr int BMt = WR * 64;
  constexpr int BNt = WC * 64;
  constexpr int BKT = 64;
  constexpr int LP = BKT + IC_LDS_PAD;
  constexpr int AKW = 4096 / BMt;    // k-width one A staging call covers
  constexpr int ACALLS = BKT / AKW;
  constexpr int BKW = 4096 / BNt;
  constexpr int BCALLS = BKT / BKW;
  __shared__ bf16raw As[BMt * LP];
  __shared__ bf16raw Bs[BNt * LP];

  const int gx = gridDim.x;
  int nwg = gx * gridDim.y;
  int orig = blockIdx.y * gx + blockIdx.x;
  int q = nwg >> 3, rr = nwg & 7;
  int wg = ((orig & 7) < rr ? (orig & 7) * (q + 1) : rr * (q + 1) + ((orig & 7) - rr) * q) +
           (orig >> 3);
  const int m0 = (wg % gx) * BMt;
  const int n0 = (wg / gx) * BNt;

  int k_begin = blockIdx.z * k_per_split;
  int k_end = min(K, k_begin + k_per_split);

  const int t = threadIdx.x;
  const int wid = t / WAVE;
  const int lane = t % WAVE;
  const int wr = wid / WC;
  const int wc = wid % WC;
  const int l15 = lane & 15, kg = lane >> 4;

  alignas(16) bf16raw rA[ACALLS * 16];
  alignas(16) bf16raw rB[BCALLS * 16];

  floatx4 acc[MFR][4] = {};

#define WG_LOAD(kt)                                                              \
  do {                                                                           \
    _Pragma("unroll") for (int ca = 0; ca < ACALLS; ++ca)                        \
        wg_stage_a<BMt>(dz, m0, CO, (kt) + ca * AKW, k_end, g, rA + ca * 16, t); \
    _Pragma("unroll") for (int cb = 0; cb < BCALLS; ++cb)                        \
        wg_stage_b<BNt>(xP, n0, N, (kt) + cb * BKW, k_end, g, rB + cb * 16, t);  \
  } while (0)

  if (k_begin < k_end) WG_LOAD(k_begin);

  for (int kt = k_begin; kt < k_end; kt += BKT) {
    if (kt > k_begin) __syncthreads();
#pragma unroll
    for (int ca = 0; ca < ACALLS; ++ca)
      wg_stage_write<BMt, LP>(As, rA + ca * 16, t, ca * AKW);
#pragma unroll
    for (int cb = 0; cb < BCALLS; ++cb)
      wg_stage_write<BNt, LP>(Bs, rB + cb * 16, t, cb * BKW);
    __syncthreads();

    if (kt + BKT < k_end) WG_LOAD(kt + BKT);

#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      cfrag_t a[MFR], b[4];
#pragma unroll
      for (int mi = 0; mi < MFR; ++mi) {
        int row = wr * 64 + mi * 16 + l15;
        a[mi] = *(const cfrag_t*)&As[row * LP + sub * 32 + ic_swz(row, kg * 8)];
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int rowb = wc * 64 + ni * 16 + l15;
        b[ni] = *(const cfrag_t*)&Bs[rowb * LP + sub * 32 + ic_swz(rowb, kg * 8)];
      }
#pragma unroll
      for (int mi = 0; mi < MFR; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[mi], b[ni], acc[mi][ni], 0, 0, 0);
    }
  }
#undef WG_LOAD

  const int m_base = m0 + wr * 64;
  const int n_base = n0 + wc * 64;
#pragma unroll
  for (int mi = 0; mi < MFR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int n = n_base + ni * 16 + l15;
      if (n >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int m = m_base + mi * 16 + kg * 4 + r;
        if (m >= CO) continue;
        if (SLAB)
          out[(int64_t)blockIdx.z * CO * N + (int64_t)m * N + n] = acc[mi][ni][r];
        else
          atomicAdd(out + (int64_t)m * N + n, acc[mi][ni][r]);
      }
    }
  }
}

// Combine slabs: dw[m,n] = sum_z ws[z,m,n] (plain write — the conv wgrad
// allocates a fresh dw, so no zero-fill pass is needed on the slab path).
__global__ void ic_wgrad_reduce_kernel(const float* __restrict__ ws, float* __restrict__ dw,
                                       int64_t MN, int zs) {
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = i0; i < MN; i += stride) {
    if (i + 4 <= MN) {
      floatx4 s = {0.f, 0.f, 0.f, 0.f};
      for (int z = 0; z < zs; ++z) {
        floatx4 v = *(const floatx4*)(ws + (int64_t)z * MN + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) s[j] += v[j];
      }
      *(floatx4*)(dw + i) = s;
    } else {
      for (int64_t k = i; k < MN; ++k) {
        float s = 0.f;
        for (int z = 0; z < zs; ++z) s += ws[(int64_t)z * MN + k];
        dw[k] = s;
      }
    }
  }
}

extern "C" hipError_t launch_conv_implicit_wgrad(const bf16raw* dz, const bf16raw* xP, float* dw,
                                                 int CO, int B, int Hp, int Wp, int CI, int KH,
                                                 int KW, int SH, int SW, int splitk, float* ws,
                                                 int OHo, int OWo, hipStream_t stream) {
  const int OH = (Hp - KH) / SH + 1, OW = (Wp - KW) / SW + 1;
  const int N = KH * KW * CI;
  const int K = B * OH * OW;
  ConvGeom g;
  g.OW = OW;
  g.OHW = OH * OW;
  g.Hp = Hp;
  g.Wp = Wp;
  g.CI = CI;
  g.KW = KW;
  g.KWCI = KW * CI;
  g.Kreal = N;
  g.SH = SH;
  g.SW = SW;
  g.OHo = OHo > 0 ? OHo : OH;
  g.OWo = OWo > 0 ? OWo : OW;
  if (splitk < 1) splitk = 1;
  int kps = K, zs = 1;
  if (splitk > 1) {
    kps = (int)ceil_div_i64(ceil_div_i64(K, splitk), 64) * 64;
    zs = (int)ceil_div_i64(K, kps);
  }
  // CO <= 32: the 64x256 flat tile (half the dead M rows, +20% MACs per
  // staged byte).  Measured: wins for CO=32 (MNIST conv2 class), slightly
  // loses at CO=64 (l1/stem class, 225->207 TF) — the 2x2 tile's extra
  // blocks apparently hide slab latency better there.
  const bool flat = (CO <= 32);
  const int bm = flat ? 64 : 128, bn = flat ? 256 : 128;
  dim3 grid((unsigned)ceil_div_i64(CO, bm), (unsigned)ceil_div_i64(N, bn), (unsigned)zs);
  if (ws != nullptr) {
    if (flat)
      // CO <= 32 -> only 2 of the 4 row fragments are live
      conv_implicit_wgrad_kernel<true, 1, 4, 2><<<grid, dim3(256), 0, stream>>>(dz, xP, ws, CO,
                                                                                N, K, g, kps);
    else
      conv_implicit_wgrad_kernel<true, 2, 2><<<grid, dim3(256), 0, stream>>>(dz, xP, ws, CO, N,
                                                                             K, g, kps);
    HIP_CHECK_LAUNCH();
    int64_t MN = (int64_t)CO * N;
    int64_t rg = ceil_div_i64(MN, 1024);
    if (rg > 2048) rg = 2048;
    ic_wgrad_reduce_kernel<<<dim3((unsigned)rg), dim3(256), 0, stream>>>(ws, dw, MN, zs);
  } else {
    if (flat)
      conv_implicit_wgrad_kernel<false, 1, 4, 2><<<grid, dim3(256), 0, stream>>>(dz, xP, dw, CO,
                                                                                 N, K, g, kps);
    else
      conv_implicit_wgrad_kernel<false, 2, 2><<<grid, dim3(256), 0, stream>>>(dz, xP, dw, CO, N,
                                                                              K, g, kps);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" int conv_implicit_wgrad_slices(int K, int splitk) {
  if (splitk < 1) splitk = 1;
  if (splitk == 1) return 1;
  int kps = (int)ceil_div_i64(ceil_div_i64(K, splitk), 64) * 64;
  return (int)ceil_div_i64(K, kps);
}
