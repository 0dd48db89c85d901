// Hand-written MFMA GEMM for gfx950 (CDNA4) — the hot op behind every
// Linear/conv forward and backward (replaces the reference's eager
// model(x)/loss.backward() CPU path, reference distributed.py:150,176).
//
// C[M,N] = A[M,K] x B[K,N]; bf16 operands (fp32 master weights are cast at
// the stage or pre-cast by the caller), fp32 MFMA accumulation
// (v_mfma_f32_16x16x32_bf16, accumulators in AGPRs).
//
// Tile variants (wave always owns a 64x64 sub-tile = 4x4 fragments):
//   <2,2>  128x128, BK=64 — the default
//   <4,1>  256x64,  BK=32 — narrow-N shapes (64-ch convs, small heads)
//   <2,4>  128x256, BK=64, 512 threads — big-N register-staged wgrad
//
// Staging (the performance-critical part; every choice here is measured —
// see profiles/bench_resnet18_optimization_log.md):
//   * k-contiguous operands (AG/BG): async global_load_lds DMA into a
//     pad-free LDS image whose 16B k-slots are XOR-swizzled by row (lane-
//     linear for the DMA, bank-spread for ds_read_b128 fragment reads);
//     boundary slabs take a vectorized swizzled fallback.
//   * transposing operands (wgrad — the reduction runs over the outer
//     stride of both matrices): T14 register pipeline — hold the slab in
//     registers, write LDS after the barrier, issue the next slab's loads
//     so they span the MFMA section.  The padded LDS image carries a
//     granule-XOR swizzle (a 16-row staging stride hits one bank group
//     whatever the row pad).
//   * double-buffered K-loop in the DMA path, single buffer + two barriers
//     in the register path.
//
// One kernel serves all three Linear/conv gradients via strides:
//   fwd   Y = X  W^T : A=X(row),   B=W^T (sbk=1,   sbn=K)    [A glds, B glds if bf16]
//   dX    = dZ W     : A=dZ(row),  B=W   (sbk=Kin, sbn=1)    [A glds]
//   dW    = dZ^T X   : A=dZ^T (sam=1, sak=N), B=X (sbk=Kin, sbn=1), split-K
//           over the batch dim with fp32 atomicAdd combine   [T14 registers]
//
// Epilogues: bias+ReLU fused into the C-write; bf16 outputs go through a
// per-wave LDS transpose so every lane stores one contiguous 16 B octet;
// the fused dW+db trick treats bias as a virtual all-ones B column
// (ones_row).  The XCD-aware block swizzle keeps an output tile's operand
// rows inside one XCD's L2.

#include "common.h"

#define BM 128
#define BN 128
#define BK 32
#define LDS_PAD 8          // 8 bf16 = 16 B: rows stay 16-byte aligned
#define LDSK (BK + LDS_PAD)

// EPI codes
#define EPI_F32 0        // fp32 store (atomicAdd when SPLITK)
#define EPI_BF16 1       // bf16 store
#define EPI_BIAS 2       // bf16 store, + bias[n]
#define EPI_BIAS_RELU 3  // bf16 store, + bias[n], relu
#define EPI_RELU 4       // bf16 store, relu (no bias)
#define EPI_F32_SLAB 5   // fp32 per-slice slab store (split-K without atomics;
                         // a separate reduce kernel combines the slabs —
                         // guide §5 "splitk-seam": slab reducer beats an
                         // fp32-atomicAdd accumulator)

typedef shortx8 frag_t;  // 8 bf16 (4 VGPRs)

// Stage a [ROWS x BK] tile (ROWS=128) into LDS laid out [ROWS][LDSK].
// Logical element (r, k) comes from src[(row0+r)*srow + (kt+k)*skol].
// 256 threads x 16 elements.  Two layouts:
//   skol==1 (k-contiguous): thread covers 16 consecutive k of one row.
//   else if srow==1 (row-contiguous): thread covers 16 consecutive rows at
//     fixed k (vectorized along rows, strided LDS column writes).
//   else: scalar element loop.
#define BF16_ONE ((bf16raw)0x3F80)

// Granule swizzle for the PADDED LDS images: XOR the 8-element (16 B)
// granule index of a 32-k half by row bits >= 4.  The srow-staging write
// pattern (16-row stride between a wave's lanes) lands every lane on the
// same bank group whatever the row pad (16-row x any-16B-aligned stride is
// 0 mod the 256 B bank period); the XOR spreads it 8-way -> ~2-way.  Lane
// groups of a fragment read share (row >> 4), so reads stay 16 B
// contiguous and conflict-free.
__device__ __forceinline__ int swz_col(int row, int col) {
  return col ^ (((row >> 4) & 3) << 3);
}

// ones_row >= 0 marks a VIRTUAL row whose every element is 1.0 (the bias
// column of dW_ext = dz^T @ [x | 1]); memory is only touched for rows below
// it.  -1 = no virtual row.
template <bool SRC_F32, int LDSTRIDE = LDSK>
__device__ __forceinline__ void stage_tile(const void* __restrict__ src, bf16raw* __restrict__ lds,
                                           int row0, int rmax, int kt, int kmax, int64_t srow,
                                           int64_t skol, int ones_row = -1) {
  const int t = threadIdx.x;
  const int mem_rows = ones_row >= 0 ? ones_row : rmax;
  if (skol == 1) {
    // 2 threads per row; 16 consecutive k each
    int r = t >> 1;
    int k0 = (t & 1) * 16;
    int gr = row0 + r;
    bf16raw* d0 = lds + r * LDSTRIDE + swz_col(r, k0);
    bf16raw* d1 = lds + r * LDSTRIDE + swz_col(r, k0 + 8);
    if (ones_row >= 0 && gr == ones_row) {
#pragma unroll
      for (int j = 0; j < 8; ++j) d0[j] = (kt + k0 + j < kmax) ? BF16_ONE : (bf16raw)0;
#pragma unroll
      for (int j = 0; j < 8; ++j) d1[j] = (kt + k0 + 8 + j < kmax) ? BF16_ONE : (bf16raw)0;
    } else if (gr < mem_rows) {
      const char* base = (const char*)src + (int64_t)gr * srow * (SRC_F32 ? 4 : 2);
      int krem = kmax - kt - k0;  // how many of our 16 k are in range
      if (krem >= 16) {
        if (SRC_F32) {
          const float* s = (const float*)base + kt + k0;
#pragma unroll
          for (int v = 0; v < 4; ++v) {
            floatx4 x = *(const floatx4*)(s + v * 4);
            bf16raw* dv = v < 2 ? d0 + v * 4 : d1 + (v - 2) * 4;
#pragma unroll
            for (int j = 0; j < 4; ++j) dv[j] = f32_to_bf16(x[j]);
          }
        } else {
          const bf16raw* s = (const bf16raw*)base + kt + k0;
          *(shortx8*)d0 = *(const shortx8*)s;
          *(shortx8*)d1 = *(const shortx8*)(s + 8);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          int k = kt + k0 + j;
          bf16raw* dj = (j < 8 ? d0 + j : d1 + (j - 8));
          if (k < kmax)
            *dj = SRC_F32 ? f32_to_bf16(((const float*)base)[k]) : ((const bf16raw*)base)[k];
          else
            *dj = 0;
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) d0[j] = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) d1[j] = 0;
    }
  } else if (srow == 1) {
    // 16 consecutive rows at fixed k; 8 threads per k column
    int k = t >> 3;           // 0..31
    int r0 = (t & 7) * 16;    // 0,16,...,112
    int gk = kt + k;
    if (gk < kmax) {
      const char* base = (const char*)src + (int64_t)gk * skol * (SRC_F32 ? 4 : 2);
      int mem_rem = mem_rows - row0 - r0;
      if (mem_rem >= 16) {
        if (SRC_F32) {
          const float* s = (const float*)base + row0 + r0;
#pragma unroll
          for (int v = 0; v < 4; ++v) {
            floatx4 x = *(const floatx4*)(s + v * 4);
#pragma unroll
            for (int j = 0; j < 4; ++j) lds[(r0 + v * 4 + j) * LDSTRIDE + swz_col(r0 + v * 4 + j, k)] = f32_to_bf16(x[j]);
          }
        } else {
          const bf16raw* s = (const bf16raw*)base + row0 + r0;
#pragma unroll
          for (int v = 0; v < 2; ++v) {
            shortx8 x = *(const shortx8*)(s + v * 8);
#pragma unroll
            for (int j = 0; j < 8; ++j) lds[(r0 + v * 8 + j) * LDSTRIDE + swz_col(r0 + v * 8 + j, k)] = (bf16raw)x[j];
          }
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          int r = r0 + j;
          bf16raw v = 0;
          if (ones_row >= 0 && row0 + r == ones_row)
            v = BF16_ONE;
          else if (j < mem_rem)  // row0 + r < mem_rows
            v = SRC_F32 ? f32_to_bf16(((const float*)base)[row0 + r])
                        : ((const bf16raw*)base)[row0 + r];
          lds[r * LDSTRIDE + swz_col(r, k)] = v;
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j) lds[(r0 + j) * LDSTRIDE + swz_col(r0 + j, k)] = 0;
    }
  } else {
    // generic scalar path
    int idx0 = t * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      int idx = idx0 + j;
      int r = idx / BK, k = idx % BK;
      int gr = row0 + r, gk = kt + k;
      bf16raw v = 0;
      if (ones_row >= 0 && gr == ones_row && gk < kmax) {
        v = BF16_ONE;
      } else if (gr < mem_rows && gk < kmax) {
        const char* p = (const char*)src + ((int64_t)gr * srow + (int64_t)gk * skol) * (SRC_F32 ? 4 : 2);
        v = SRC_F32 ? f32_to_bf16(*(const float*)p) : *(const bf16raw*)p;
      }
      lds[r * LDSTRIDE + swz_col(r, k)] = v;
    }
  }
}

// Split staging (guide T14): stage_load fills 16 per-thread registers for a
// 32-k chunk; stage_write stores them to the padded LDS image.  Holding the
// slab in registers lets the global loads for slab t+1 be issued right
// after the barrier and span the whole MFMA section of slab t, instead of
// stalling on vmcnt just before it — this is the latency fix for the wgrad
// shapes where neither operand is k-contiguous (no glds possible).
template <bool SRC_F32>
__device__ __forceinline__ void stage_load(const void* __restrict__ src, int row0, int rmax,
                                           int kt, int kmax, int64_t srow, int64_t skol,
                                           int ones_row, bf16raw* __restrict__ regs,
                                           int t) {
  const int mem_rows = ones_row >= 0 ? ones_row : rmax;
  if (skol == 1) {
    int r = t >> 1;
    int k0 = (t & 1) * 16;
    int gr = row0 + r;
    if (ones_row >= 0 && gr == ones_row) {
#pragma unroll
      for (int j = 0; j < 16; ++j) regs[j] = (kt + k0 + j < kmax) ? BF16_ONE : (bf16raw)0;
    } else if (gr < mem_rows) {
      const char* base = (const char*)src + (int64_t)gr * srow * (SRC_F32 ? 4 : 2);
      int krem = kmax - kt - k0;
      if (krem >= 16) {
        if (SRC_F32) {
          const float* sp = (const float*)base + kt + k0;
#pragma unroll
          for (int v = 0; v < 4; ++v) {
            floatx4 x = *(const floatx4*)(sp + v * 4);
#pragma unroll
            for (int j = 0; j < 4; ++j) regs[v * 4 + j] = f32_to_bf16(x[j]);
          }
        } else {
          const bf16raw* sp = (const bf16raw*)base + kt + k0;
          *(shortx8*)regs = *(const shortx8*)sp;
          *(shortx8*)(regs + 8) = *(const shortx8*)(sp + 8);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          int k = kt + k0 + j;
          regs[j] = (k < kmax)
                        ? (SRC_F32 ? f32_to_bf16(((const float*)base)[k]) : ((const bf16raw*)base)[k])
                        : (bf16raw)0;
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j) regs[j] = 0;
    }
  } else if (srow == 1) {
    int k = t >> 3;
    int r0 = (t & 7) * 16;
    int gk = kt + k;
    if (gk < kmax) {
      const char* base = (const char*)src + (int64_t)gk * skol * (SRC_F32 ? 4 : 2);
      int mem_rem = mem_rows - row0 - r0;
      if (mem_rem >= 16) {
        if (SRC_F32) {
          const float* sp = (const float*)base + row0 + r0;
#pragma unroll
          for (int v = 0; v < 4; ++v) {
            floatx4 x = *(const floatx4*)(sp + v * 4);
#pragma unroll
            for (int j = 0; j < 4; ++j) regs[v * 4 + j] = f32_to_bf16(x[j]);
          }
        } else {
          const bf16raw* sp = (const bf16raw*)base + row0 + r0;
          *(shortx8*)regs = *(const shortx8*)sp;
          *(shortx8*)(regs + 8) = *(const shortx8*)(sp + 8);
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          bf16raw v = 0;
          if (ones_row >= 0 && row0 + r0 + j == ones_row)
            v = BF16_ONE;
          else if (j < mem_rem)
            v = SRC_F32 ? f32_to_bf16(((const float*)base)[row0 + r0 + j])
                        : ((const bf16raw*)base)[row0 + r0 + j];
          regs[j] = v;
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j) regs[j] = 0;
    }
  } else {
    int idx0 = t * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      int idx = idx0 + j;
      int r = idx / BK, k = idx % BK;
      int gr = row0 + r, gk = kt + k;
      bf16raw v = 0;
      if (ones_row >= 0 && gr == ones_row && gk < kmax) {
        v = BF16_ONE;
      } else if (gr < mem_rows && gk < kmax) {
        const char* p =
            (const char*)src + ((int64_t)gr * srow + (int64_t)gk * skol) * (SRC_F32 ? 4 : 2);
        v = SRC_F32 ? f32_to_bf16(*(const float*)p) : *(const bf16raw*)p;
      }
      regs[j] = v;
    }
  }
}

template <int LDSTRIDE>
__device__ __forceinline__ void stage_write(bf16raw* __restrict__ lds, int64_t srow,
                                            int64_t skol, const bf16raw* __restrict__ regs,
                                            int t) {
  if (skol == 1) {
    int r = t >> 1;
    int k0 = (t & 1) * 16;
    *(shortx8*)(lds + r * LDSTRIDE + swz_col(r, k0)) = *(const shortx8*)regs;
    *(shortx8*)(lds + r * LDSTRIDE + swz_col(r, k0 + 8)) = *(const shortx8*)(regs + 8);
  } else if (srow == 1) {
    int k = t >> 3;
    int r0 = (t & 7) * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) lds[(r0 + j) * LDSTRIDE + swz_col(r0 + j, k)] = regs[j];
  } else {
    int idx0 = t * 16;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      int idx = idx0 + j;
      lds[(idx / BK) * LDSTRIDE + swz_col(idx / BK, idx % BK)] = regs[j];
    }
  }
}

// ---------------------------------------------------------------------------
// Direct global->LDS staging (glds) for the k-contiguous A operand.  The LDS
// image is pad-free [ROWS][BKT] bf16 with the 16-byte k-slot of each row
// XOR-swizzled by (row & (SLOTS-1)) — lane-linear for the DMA (dest is
// wave-uniform base + lane*16; guide §5 "Async global->LDS"), bank-spread
// for the ds_read_b128 fragment reads.  The swizzle permutes 16B pieces
// within one 128B line of the source row, so global coalescing is kept.
// ---------------------------------------------------------------------------

template <int ROWS, int BKT>
__device__ __forceinline__ void stage_glds(const bf16raw* __restrict__ src,
                                           bf16raw* __restrict__ lds, int row0, int64_t sam,
                                           int kt) {
  constexpr int SLOTS = BKT / 8;       // 16B slots per LDS row
  constexpr int CH_ROWS = 64 / SLOTS;  // rows per 1KB wave chunk
  constexpr int NCHUNK = ROWS / CH_ROWS;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int r_in = lane / SLOTS;
  const int slot = lane % SLOTS;
#pragma unroll
  for (int c = wid; c < NCHUNK; c += 4) {
    int row = c * CH_ROWS + r_in;
    int sslot = slot ^ (row & (SLOTS - 1));
    const bf16raw* g = src + (int64_t)(row0 + row) * sam + kt + sslot * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(lds + c * CH_ROWS * BKT), 16, 0, 0);
  }
}

// boundary-slab fallback writing the SAME swizzled image with zero fill.
// Vectorized: AG implies a k-contiguous source, so full granules move as
// shortx8 (the K<BKT degenerate shapes — e.g. a K=32 dgrad — run this every
// slab, and the original per-element version was the whole kernel's cost).
template <int ROWS, int BKT>
__device__ __forceinline__ void stage_swz_fallback(const bf16raw* __restrict__ src,
                                                   bf16raw* __restrict__ lds, int row0, int rmax,
                                                   int kt, int kmax, int64_t sam) {
  constexpr int SLOTS = BKT / 8;
  constexpr int TPR = 256 / ROWS;    // threads per row (1 or 2)
  constexpr int KSPAN = BKT / TPR;   // k elements per thread
  const int t = threadIdx.x;
  const int r = t / TPR;
  const int kb = (t % TPR) * KSPAN;
  const int gr = row0 + r;
  const bf16raw* base = src + (int64_t)gr * sam + kt;
#pragma unroll
  for (int g = 0; g < KSPAN / 8; ++g) {
    int k = kb + g * 8;
    bf16raw* dst = lds + r * BKT + (((k >> 3) ^ (r & (SLOTS - 1))) << 3);
    if (gr < rmax && kt + k + 8 <= kmax) {
      *(shortx8*)dst = *(const shortx8*)(base + k);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dst[j] = (gr < rmax && kt + k + j < kmax) ? base[k + j] : (bf16raw)0;
    }
  }
}

// Wave arrangement WR x WC (each wave owns a 64x64 sub-tile = 4x4 MFMA
// fragments): <2,2> = 128x128 block tile with BK=64, <4,1> = 256x64 with
// BK=32 for narrow-N shapes.  The A operand (the big streamed matrix) is
// staged by async global_load_lds DMA into a pad-free XOR-swizzled image
// when it is k-contiguous (AG) — the guide's ladder measured 517->874 TF
// from exactly this change on the 128^2-tile structure — and by the padded
// register path otherwise (wgrad, where the reduction runs over the outer
// stride of both operands).  B (small, L2-hot, possibly fp32 master
// weights) always uses the register path.  K-loop is double-buffered.
template <bool B_IS_F32, int EPI, bool SPLITK, int WR, int WC, bool AG, bool BG = false>
__global__ __launch_bounds__(WR * WC * 64, 2) void gemm_kernel(const void* __restrict__ Ap,
                                                   const void* __restrict__ Bp,
                                                   float* __restrict__ Cf,
                                                   bf16raw* __restrict__ Cb,
                                                   const float* __restrict__ bias, int M, int N,
                                                   int K, int64_t sam, int64_t sak, int64_t sbk,
                                                   int64_t sbn, int k_per_split,
                                                   float* __restrict__ Db, int ones_row) {
  constexpr int BMt = WR * 64;
  constexpr int BNt = WC * 64;
  constexpr int BKT = (WC == 1) ? 32 : 64;  // narrow tile keeps BK=32 (LDS budget)
  constexpr int SUBS = BKT / 32;            // MFMA k-steps per slab
  constexpr int SLOTS = BKT / 8;
  constexpr int LP = BKT + LDS_PAD;         // padded stride (register-staged images)
  constexpr int BROWS = BNt < 128 ? 128 : BNt;  // stage_tile writes 128 LDS rows
  constexpr int NBUF = AG ? 2 : 1;  // T14 register pipeline needs one buffer
  __shared__ bf16raw As[NBUF][AG ? (BMt * BKT) : (BMt * LP)];
  __shared__ bf16raw Bs[NBUF][BG ? (BROWS * BKT) : (BROWS * LP)];

  // bijective XCD-aware swizzle of the flattened block id (guide §5.5 T1):
  // consecutive output tiles land on one XCD so shared operand rows stay in
  // that XCD's L2.
  const int gx = gridDim.x;
  int nwg = gx * gridDim.y;
  int orig = blockIdx.y * gx + blockIdx.x;
  int q = nwg >> 3, rr = nwg & 7;
  int wg = ((orig & 7) < rr ? (orig & 7) * (q + 1) : rr * (q + 1) + ((orig & 7) - rr) * q) +
           (orig >> 3);
  const int m0 = (wg % gx) * BMt;
  const int n0 = (wg / gx) * BNt;

  int k_begin = 0, k_end = K;
  if (SPLITK) {
    k_begin = blockIdx.z * k_per_split;
    k_end = min(K, k_begin + k_per_split);
  }

  constexpr int THREADS = WR * WC * 64;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wr = wid / WC;
  const int wc = wid % WC;
  const int l15 = lane & 15, kg = lane >> 4;

  const bool a_rows_full = (m0 + BMt <= M);

  floatx4 acc[4][4] = {};

  if constexpr (AG && THREADS == 256) {
    // A by async DMA (double-buffered prefetch), B by direct register stage.
#define STAGE_SLAB(bufi, kt)                                                               \
  do {                                                                                     \
    if (a_rows_full && (kt) + BKT <= k_end)                                                \
      stage_glds<BMt, BKT>((const bf16raw*)Ap, As[bufi], m0, sam, kt);                     \
    else                                                                                   \
      stage_swz_fallback<BMt, BKT>((const bf16raw*)Ap, As[bufi], m0, M, kt, k_end, sam);   \
    /* B staged TRANSPOSED: LDS row = n, col = k -> srow := sbn, skol := sbk */            \
    if (BG) {                                                                              \
      if (n0 + BROWS <= N && (kt) + BKT <= k_end)                                          \
        stage_glds<BROWS, BKT>((const bf16raw*)Bp, Bs[1 && (bufi)], n0, sbn, kt);          \
      else                                                                                 \
        stage_swz_fallback<BROWS, BKT>((const bf16raw*)Bp, Bs[1 && (bufi)], n0, N, kt,     \
                                       k_end, sbn);                                        \
    } else {                                                                               \
      _Pragma("unroll") for (int kh = 0; kh < BKT; kh += 32)                               \
          stage_tile<B_IS_F32, LP>(Bp, Bs[1 && (bufi)] + kh, n0, N, (kt) + kh, k_end,      \
                                   sbn, sbk, ones_row);                                    \
    }                                                                                      \
  } while (0)

    int buf = 0;
    if (k_begin < k_end) STAGE_SLAB(0, k_begin);
    __syncthreads();

    for (int kt = k_begin; kt < k_end; kt += BKT) {
      if (kt + BKT < k_end) STAGE_SLAB(buf ^ 1, kt + BKT);  // prefetch next slab

#pragma unroll
      for (int sub = 0; sub < SUBS; ++sub) {
        frag_t a[4], b[4];
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          int row = wr * 64 + mi * 16 + l15;
          int kq = kg + sub * 4;
          a[mi] = *(const frag_t*)&As[buf][row * BKT + (((kq) ^ (row & (SLOTS - 1))) << 3)];
        }
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          int rowb = wc * 64 + ni * 16 + l15;
          int kq = kg + sub * 4;
          b[ni] = BG ? *(const frag_t*)&Bs[buf][rowb * BKT + (((kq) ^ (rowb & (SLOTS - 1))) << 3)]
                     : *(const frag_t*)&Bs[buf][rowb * LP + sub * 32 + swz_col(rowb, kg * 8)];
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[mi][ni] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[mi], b[ni], acc[mi][ni], 0, 0, 0);
      }

      __syncthreads();
      buf ^= 1;
    }
#undef STAGE_SLAB
  } else {
    // T14 register pipeline, single LDS buffer: hold slab t in registers,
    // write it to LDS after the barrier, immediately issue slab t+1's
    // loads, then run slab t's MFMAs while those loads fly.  At 512
    // threads (the 128x256 wgrad tile) the two thread-halves split the
    // work: half g stages A k-half g and B row-half g (both k-halves).
    constexpr int NG = THREADS / 256;                  // 1 or 2
    constexpr int AH = (BMt / 128) * (BKT / 32) / NG;  // A chunks per group
    constexpr int BH = (BNt < 128 ? 128 : BNt) / 128 * (BKT / 32) / NG;
    alignas(16) bf16raw rA[AH * 16];
    alignas(16) bf16raw rB[BH * 16];
    const int grp = NG == 1 ? 0 : (threadIdx.x >> 8);
    const int gt = NG == 1 ? threadIdx.x : (threadIdx.x & 255);

#define LOAD_SLAB(kt)                                                                      \
  do {                                                                                     \
    if (NG == 1) {                                                                         \
      _Pragma("unroll") for (int ro = 0; ro < BMt / 128; ++ro) {                           \
        _Pragma("unroll") for (int kh = 0; kh < BKT / 32; ++kh)                            \
            stage_load<false>(Ap, m0 + ro * 128, M, (kt) + kh * 32, k_end, sam, sak, -1,   \
                              rA + (ro * (BKT / 32) + kh) * 16, gt);                       \
      }                                                                                    \
      _Pragma("unroll") for (int kh = 0; kh < BKT / 32; ++kh)                              \
          stage_load<B_IS_F32>(Bp, n0, N, (kt) + kh * 32, k_end, sbn, sbk, ones_row,       \
                               rB + kh * 16, gt);                                          \
    } else {                                                                               \
      /* group g: A 128x32 k-half g; B row-half (n0+g*128) both k-halves */                \
      stage_load<false>(Ap, m0, M, (kt) + grp * 32, k_end, sam, sak, -1, rA, gt);          \
      _Pragma("unroll") for (int kh = 0; kh < BKT / 32; ++kh)                              \
          stage_load<B_IS_F32>(Bp, n0 + grp * 128, N, (kt) + kh * 32, k_end, sbn, sbk,     \
                               ones_row, rB + kh * 16, gt);                                \
    }                                                                                      \
  } while (0)

    if (k_begin < k_end) LOAD_SLAB(k_begin);

    for (int kt = k_begin; kt < k_end; kt += BKT) {
      if (kt > k_begin) __syncthreads();  // prior slab's MFMA reads done
      if (NG == 1) {
#pragma unroll
        for (int ro = 0; ro < BMt / 128; ++ro) {
#pragma unroll
          for (int kh = 0; kh < BKT / 32; ++kh)
            stage_write<LP>(As[0] + ro * 128 * LP + kh * 32, sam, sak,
                            rA + (ro * (BKT / 32) + kh) * 16, gt);
        }
#pragma unroll
        for (int kh = 0; kh < BKT / 32; ++kh)
          stage_write<LP>(Bs[0] + kh * 32, sbn, sbk, rB + kh * 16, gt);
      } else {
        stage_write<LP>(As[0] + grp * 32, sam, sak, rA, gt);
#pragma unroll
        for (int kh = 0; kh < BKT / 32; ++kh)
          stage_write<LP>(Bs[0] + grp * 128 * LP + kh * 32, sbn, sbk, rB + kh * 16, gt);
      }
      __syncthreads();  // publish

      if (kt + BKT < k_end) LOAD_SLAB(kt + BKT);  // loads span the MFMAs below

#pragma unroll
      for (int sub = 0; sub < SUBS; ++sub) {
        frag_t a[4], b[4];
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          int row = wr * 64 + mi * 16 + l15;
          a[mi] = *(const frag_t*)&As[0][row * LP + sub * 32 + swz_col(row, kg * 8)];
        }
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          int rowb = wc * 64 + ni * 16 + l15;
          b[ni] = *(const frag_t*)&Bs[0][rowb * LP + sub * 32 + swz_col(rowb, kg * 8)];
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[mi][ni] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[mi], b[ni], acc[mi][ni], 0, 0, 0);
      }
    }
#undef LOAD_SLAB
  }

  // epilogue.  bf16 outputs go through a per-wave LDS transpose so each
  // lane stores one contiguous 16 B octet (scalar 2 B stores measured the
  // C-write path at ~2.3 TB/s vs ~4.4 for reads); the MFMA fragment layout
  // (col = lane&15, row = 4*(lane>>4)+reg) cannot produce contiguous
  // per-lane stores directly.  fp32/split-K keeps the scalar path (atomics).
  const int m_base = m0 + wr * 64;
  const int n_base = n0 + wc * 64;
  if (EPI != EPI_F32 && EPI != EPI_F32_SLAB && (N & 7) == 0 && n_base + 64 <= N) {
    // all waves must be past their final fragment reads before the scratch
    // overwrites the staging images (the T14 loop has no trailing barrier)
    __syncthreads();
    // per-wave [16][68] fp32 scratch inside the (now idle) LDS staging
    // buffers: As holds 4x(16x68x4B) for the 4-wave tiles; the 8-wave wide
    // tile needs 34.8 KB and uses Bs (36.9 KB there)
    constexpr int EPAD = 68;
    float* ep = (THREADS == 512 ? (float*)&Bs[0][0] : (float*)&As[0][0]) + wid * 16 * EPAD;
    const int orow = lane >> 2;          // 0..15 output row of the mi-slice
    const int oct = lane & 3;            // which 16-col half-octet pair
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) ep[(kg * 4 + r) * EPAD + ni * 16 + l15] = acc[mi][ni][r];
      __builtin_amdgcn_s_waitcnt(0);  // lgkm: own-wave LDS writes visible
      int m = m_base + mi * 16 + orow;
      if (m < M) {
#pragma unroll
        for (int h = 0; h < 2; ++h) {
          int c0 = oct * 16 + h * 8;
          int n = n_base + c0;
          alignas(16) short outp[8];
          const float* src = ep + orow * EPAD + c0;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float val = src[j];
            if (EPI == EPI_BIAS || EPI == EPI_BIAS_RELU) val += bias[n + j];
            if (EPI == EPI_BIAS_RELU || EPI == EPI_RELU) val = fmaxf(val, 0.f);
            outp[j] = (short)f32_to_bf16(val);
          }
          *(shortx8*)(Cb + (int64_t)m * N + n) = *(const shortx8*)outp;
        }
      }
      __builtin_amdgcn_s_waitcnt(0);  // all lanes done reading before reuse
    }
    return;
  }
  if (EPI == EPI_F32_SLAB) {
    // split-K slab: this slice's full [BMtxBNt] fp32 tile goes to its own
    // workspace slab with PLAIN coalesced stores (consecutive lanes hit
    // consecutive n) — no fabric-serialized per-dword atomics.  The ones
    // column rides along at its n; the reduce kernel routes it to db.
    float* ws = Cf + (int64_t)blockIdx.z * M * N;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int n = n_base + ni * 16 + l15;
        if (n >= N) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int m = m_base + mi * 16 + kg * 4 + r;
          if (m >= M) continue;
          ws[(int64_t)m * N + n] = acc[mi][ni][r];
        }
      }
    }
    return;
  }
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int n = n_base + ni * 16 + l15;
      if (n >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int m = m_base + mi * 16 + kg * 4 + r;
        if (m >= M) continue;
        float val = acc[mi][ni][r];
        if (EPI == EPI_F32) {
          if (Db != nullptr && n == ones_row) {
            atomicAdd(Db + m, val);  // bias column of dW_ext
          } else if (SPLITK) {
            atomicAdd(Cf + (int64_t)m * (N - (ones_row >= 0 ? 1 : 0)) + n, val);
          } else {
            Cf[(int64_t)m * (N - (ones_row >= 0 ? 1 : 0)) + n] = val;
          }
        } else {
          int64_t off = (int64_t)m * N + n;
          if (EPI == EPI_BIAS || EPI == EPI_BIAS_RELU) val += bias[n];
          if (EPI == EPI_BIAS_RELU || EPI == EPI_RELU) val = fmaxf(val, 0.f);
          Cb[off] = f32_to_bf16(val);
        }
      }
    }
  }
}

// Combine the split-K slabs: dw[m,n] += sum_z ws[z,m,n]; the virtual ones
// column (bias) routes to db.  Memory-bound: zs*M*N fp32 reads, float4 loads
// where the row layout allows.
// TRANSPOSE=false: dw is [M][N-ones] row-major (+ db for the ones column).
// TRANSPOSE=true: the GEMM computed dW^T (slabs are [M=Kcol][N=CO]); store
// dw[n][m] — the transposed-wgrad route for CO<=64 layers where the direct
// orientation wastes half a 128-row tile (M utilization).  The scattered
// 4 B writes are MN elements of a tiny matrix; the coalesced zs*MN slab
// reads dominate.
template <bool TRANSPOSE>
__global__ void wgrad_reduce_kernel(const float* __restrict__ ws, float* __restrict__ dw,
                                    float* __restrict__ db, int64_t MN, int N, int zs,
                                    int ones_row) {
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  const int nw = N - (ones_row >= 0 ? 1 : 0);
  const int64_t Mrows = MN / N;
  for (int64_t i = i0; i < MN; i += stride) {
    if (i + 4 <= MN && (i % N) + 4 <= (int64_t)N) {
      floatx4 s = {0.f, 0.f, 0.f, 0.f};
      for (int z = 0; z < zs; ++z) {
        floatx4 v = *(const floatx4*)(ws + (int64_t)z * MN + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) s[j] += v[j];
      }
      int n = (int)(i % N);
      int64_t m = i / N;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        if (TRANSPOSE) {
          dw[(int64_t)(n + j) * Mrows + m] += s[j];
        } else if (ones_row >= 0 && n + j == ones_row) {
          if (db) db[m] += s[j];
        } else {
          dw[m * nw + n + j] += s[j];
        }
      }
    } else {
      for (int64_t k = i; k < i + 4 && k < MN; ++k) {
        float s = 0.f;
        for (int z = 0; z < zs; ++z) s += ws[(int64_t)z * MN + k];
        int n = (int)(k % N);
        int64_t m = k / N;
        if (TRANSPOSE) {
          dw[(int64_t)n * Mrows + m] += s;
        } else if (ones_row >= 0 && n == ones_row) {
          if (db) db[m] += s;
        } else {
          dw[m * nw + n] += s;
        }
      }
    }
  }
}

// Host helper: number of K slices the slab path will use for (K, splitk) —
// callers size the workspace as zs * M * N floats.
extern "C" int wgrad_slab_slices(int K, int splitk) {
  if (splitk < 0) splitk = -splitk;
  if (splitk < 1) splitk = 1;
  if (splitk == 1) return 1;
  int kps = (int)ceil_div_i64(ceil_div_i64(K, splitk), 64) * 64;
  return (int)ceil_div_i64(K, kps);
}

// Split-K wgrad without atomics: per-slice slab stores + one reduce launch.
// Same operand contract as launch_gemm_bf16's dW path (A = dz^T k-strided,
// B = x k-strided, optional virtual ones column for the fused bias grad);
// ws must hold wgrad_slab_slices(K, splitk) * M * N floats.
// transpose != 0: the caller swapped operands to compute dW^T (M = Kcol,
// N = CO); the reduce writes dw back in [CO][Kcol] orientation.  Use for
// CO <= 64 layers (stem/l1-class) where direct orientation leaves half the
// 128-row tile dead.  ones_row must be -1 with transpose.
extern "C" hipError_t launch_wgrad_slab(const void* A, const void* B, int b_is_f32, float* dw,
                                        float* db, int M, int N, int K, int64_t sam, int64_t sak,
                                        int64_t sbk, int64_t sbn, int splitk, int ones_row,
                                        float* ws, int transpose, hipStream_t stream) {
  if (splitk < 0) splitk = -splitk;
  if (splitk < 1) splitk = 1;
  int kps = K;
  int zs = 1;
  if (splitk > 1) {
    kps = (int)ceil_div_i64(ceil_div_i64(K, splitk), 64) * 64;
    zs = (int)ceil_div_i64(K, kps);
  }
  const bool narrow = (N <= 64 && (transpose ? M > 64 : M > 256));
  const bool wide = !narrow && N >= 256 && M > 64;
  const int bm = narrow ? 256 : BM, bn = narrow ? 64 : (wide ? 256 : BN);
  dim3 grid((unsigned)ceil_div_i64(M, bm), (unsigned)ceil_div_i64(N, bn), (unsigned)zs);
  dim3 block(wide ? 512 : 256);

#define DISPATCH_SLAB(BF32)                                                                   \
  do {                                                                                        \
    if (wide)                                                                                 \
      gemm_kernel<BF32, EPI_F32_SLAB, true, 2, 4, false><<<grid, block, 0, stream>>>(         \
          A, B, ws, nullptr, nullptr, M, N, K, sam, sak, sbk, sbn, kps, nullptr, ones_row); \
    else if (narrow)                                                                          \
      gemm_kernel<BF32, EPI_F32_SLAB, true, 4, 1, false><<<grid, block, 0, stream>>>(         \
          A, B, ws, nullptr, nullptr, M, N, K, sam, sak, sbk, sbn, kps, nullptr, ones_row); \
    else                                                                                      \
      gemm_kernel<BF32, EPI_F32_SLAB, true, 2, 2, false><<<grid, block, 0, stream>>>(         \
          A, B, ws, nullptr, nullptr, M, N, K, sam, sak, sbk, sbn, kps, nullptr, ones_row); \
  } while (0)

  if (b_is_f32)
    DISPATCH_SLAB(true);
  else
    DISPATCH_SLAB(false);
#undef DISPATCH_SLAB
  HIP_CHECK_LAUNCH();

  int64_t MN = (int64_t)M * N;
  int rblock = 256;
  int64_t rg = ceil_div_i64(MN, (int64_t)rblock * 4);
  if (rg > 2048) rg = 2048;
  if (transpose)
    wgrad_reduce_kernel<true><<<dim3((unsigned)rg), dim3(rblock), 0, stream>>>(ws, dw, db, MN, N,
                                                                               zs, ones_row);
  else
    wgrad_reduce_kernel<false><<<dim3((unsigned)rg), dim3(rblock), 0, stream>>>(ws, dw, db, MN, N,
                                                                                zs, ones_row);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

extern "C" hipError_t launch_gemm_bf16(const void* A, const void* B, int b_is_f32, float* Cf,
                                       bf16raw* Cb, const float* bias, int M, int N, int K,
                                       int64_t sam, int64_t sak, int64_t sbk, int64_t sbn, int epi,
                                       int splitk, float* Db, int ones_row, hipStream_t stream) {
  dim3 block(256);
  // splitk < 0 => |splitk| slices AND force the atomic (accumulate) epilogue
  // even if the recomputed slice count collapses to 1.
  int force_atomic = 0;
  if (splitk < 0) {
    force_atomic = 1;
    splitk = -splitk;
  }
  int kps = 0;
  if (splitk < 1) splitk = 1;
  if (splitk > 1) {
    // slices are multiples of 64 so both BK=64 and BK=32 variants align
    kps = (int)ceil_div_i64(ceil_div_i64(K, splitk), 64) * 64;
    splitk = (int)ceil_div_i64(K, kps);
  }
  if (splitk == 1) kps = K;
  const bool atomic = (splitk > 1 || force_atomic);
  // 256x64 (4x1-wave) tiles win for narrow N once the kernel is held to
  // 2 waves/SIMD (tools/gemm_bench.hip: stem fwd 686us vs 754us, l1 conv
  // 348us vs 388us)
  const bool narrow = (N <= 64 && M > 256);
  // A is DMA-staged (glds) when k-contiguous with 16B-aligned rows
  const bool ag0 = (sak == 1) && (sam % 8 == 0);
  // 128x256 (2x4-wave, 512-thread) tiles for the register-staged big-N
  // shapes (wgrad): halves the A re-reads, doubles MACs per staged byte
  const bool wide = !narrow && !ag0 && N >= 256 && M > 64;
  const int bm = narrow ? 256 : BM, bn = narrow ? 64 : (wide ? 256 : BN);
  dim3 grid((unsigned)ceil_div_i64(M, bm), (unsigned)ceil_div_i64(N, bn), (unsigned)splitk);
  if (wide) block = dim3(512);

  const bool ag = ag0;
  // B by DMA too when it is a bf16 k-contiguous operand (conv fwd with the
  // pre-cast w2d) — the guide's step-3 ladder result (517->874 TF) stages
  // BOTH operands by glds
  const bool bg = (!b_is_f32) && (sbk == 1) && (sbn % 8 == 0) && ones_row < 0;

#define DISPATCH(BF32, EPIC, SPK)                                                            \
  do {                                                                                       \
    if (wide)                                                                                \
      gemm_kernel<BF32, EPIC, SPK, 2, 4, false><<<grid, block, 0, stream>>>(                 \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
    else if (narrow && ag && bg)                                                             \
      gemm_kernel<BF32, EPIC, SPK, 4, 1, true, true><<<grid, block, 0, stream>>>(            \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
    else if (narrow && ag)                                                                   \
      gemm_kernel<BF32, EPIC, SPK, 4, 1, true><<<grid, block, 0, stream>>>(                  \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
    else if (narrow)                                                                         \
      gemm_kernel<BF32, EPIC, SPK, 4, 1, false><<<grid, block, 0, stream>>>(                 \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
    else if (ag && bg)                                                                       \
      gemm_kernel<BF32, EPIC, SPK, 2, 2, true, true><<<grid, block, 0, stream>>>(            \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
    else if (ag)                                                                             \
      gemm_kernel<BF32, EPIC, SPK, 2, 2, true><<<grid, block, 0, stream>>>(                  \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
    else                                                                                     \
      gemm_kernel<BF32, EPIC, SPK, 2, 2, false><<<grid, block, 0, stream>>>(                 \
          A, B, Cf, Cb, bias, M, N, K, sam, sak, sbk, sbn, kps, Db, ones_row);               \
  } while (0)

  if (b_is_f32) {
    if (atomic && epi == EPI_F32) DISPATCH(true, EPI_F32, true);
    else if (epi == EPI_F32) DISPATCH(true, EPI_F32, false);
    else if (epi == EPI_BF16) DISPATCH(true, EPI_BF16, false);
    else if (epi == EPI_BIAS) DISPATCH(true, EPI_BIAS, false);
    else if (epi == EPI_BIAS_RELU) DISPATCH(true, EPI_BIAS_RELU, false);
    else if (epi == EPI_RELU) DISPATCH(true, EPI_RELU, false);
    else return hipErrorInvalidValue;
  } else {
    if (atomic && epi == EPI_F32) DISPATCH(false, EPI_F32, true);
    else if (epi == EPI_F32) DISPATCH(false, EPI_F32, false);
    else if (epi == EPI_BF16) DISPATCH(false, EPI_BF16, false);
    else if (epi == EPI_BIAS) DISPATCH(false, EPI_BIAS, false);
    else if (epi == EPI_BIAS_RELU) DISPATCH(false, EPI_BIAS_RELU, false);
    else if (epi == EPI_RELU) DISPATCH(false, EPI_RELU, false);
    else return hipErrorInvalidValue;
  }
#undef DISPATCH
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}
