// Python bindings for the sparktorch_amd gfx950 kernels (_sparkhip).
// Host-only translation unit: tensor validation + launcher calls on the
// current HIP stream.  Device code lives in the .hip files.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdint>

typedef uint16_t bf16raw;

extern "C" {
hipError_t launch_fused_adam(float*, const float*, float*, float*, int64_t, float, float, float,
                             float, float, float, float, float, int, const int*, hipStream_t);
hipError_t launch_increment_i32(int*, hipStream_t);
hipError_t launch_fused_sgd(float*, const float*, float*, int64_t, float, float, float, float,
                            float, int, int, int, hipStream_t);
hipError_t launch_relu_bwd(const bf16raw*, const bf16raw*, bf16raw*, int64_t, hipStream_t);
hipError_t launch_bias_grad(const bf16raw*, float*, int, int, float*, int,
                            hipStream_t);
hipError_t launch_small_wgrad(const bf16raw*, const bf16raw*, float*, float*, int, int64_t, int,
                              int, hipStream_t);
hipError_t launch_pad_nhwc(const bf16raw*, bf16raw*, int, int, int, int, int, hipStream_t);
hipError_t launch_s2d_stem(const bf16raw*, bf16raw*, int, int, int, int, hipStream_t);
hipError_t launch_flip_w2d(const bf16raw*, bf16raw*, int, int, int, hipStream_t);
hipError_t launch_conv_implicit_fwd(const bf16raw*, const bf16raw*, const float*, bf16raw*, int,
                                    int, int, int, int, int, int, int, int, int, int, int, int,
                                    hipStream_t);
hipError_t launch_conv_implicit_wgrad(const bf16raw*, const bf16raw*, float*, int, int, int, int,
                                      int, int, int, int, int, int, float*, int, int,
                                      hipStream_t);
int conv_implicit_wgrad_slices(int, int);
hipError_t launch_cast_f64_f32(const double*, float*, int64_t, hipStream_t);
hipError_t launch_cast_f64_bf16(const double*, bf16raw*, int64_t, hipStream_t);
hipError_t launch_cast_f32_bf16(const float*, bf16raw*, int64_t, hipStream_t);
hipError_t launch_ce_fused(const bf16raw*, const int64_t*, float*, bf16raw*, int, int,
                           hipStream_t);
hipError_t launch_mse_fused(const bf16raw*, const bf16raw*, float*, bf16raw*, int64_t,
                            hipStream_t);
hipError_t launch_gemm_bf16(const void*, const void*, int, float*, bf16raw*, const float*, int,
                            int, int, int64_t, int64_t, int64_t, int64_t, int, int, float*, int,
                            hipStream_t);
hipError_t launch_bn_stats(const bf16raw*, float*, float*, float*, float*, float*, float*, int,
                           int, int64_t, float, float, int, hipStream_t);
hipError_t launch_bn_apply(const bf16raw*, const bf16raw*, bf16raw*, const float*, const float*,
                           const float*, const float*, int, int64_t, int64_t, int, hipStream_t);
hipError_t launch_bn_bwd_reduce(const bf16raw*, const bf16raw*, const bf16raw*, const float*,
                                const float*, float*, float*, int, int, int64_t, int,
                                hipStream_t);
hipError_t launch_bn_bwd_dx(const bf16raw*, const bf16raw*, const bf16raw*, const float*,
                            const float*, const float*, const float*, const float*, bf16raw*,
                            int, int64_t, int64_t, float, int, hipStream_t);
hipError_t launch_add_relu(const bf16raw*, const bf16raw*, bf16raw*, int64_t, hipStream_t);
hipError_t launch_maxpool_gen_fwd(const bf16raw*, bf16raw*, uint8_t*, int64_t, int, int, int,
                                  int, int, int, int, hipStream_t);
hipError_t launch_maxpool_gen_bwd(const bf16raw*, const uint8_t*, bf16raw*, int64_t, int, int,
                                  int, int, int, int, int, hipStream_t);
hipError_t launch_gap_fwd(const bf16raw*, bf16raw*, int64_t, int64_t, hipStream_t);
hipError_t launch_gap_bwd(const bf16raw*, bf16raw*, int64_t, int64_t, hipStream_t);
hipError_t launch_im2col_nhwc(const bf16raw*, bf16raw*, int, int, int, int, int, int, int, int,
                              int, int, int, int, int, hipStream_t);
hipError_t launch_col2im_nhwc(const bf16raw*, bf16raw*, int, int, int, int, int, int, int, int,
                              int, int, int, int, int, hipStream_t);
hipError_t launch_maxpool_nhwc_fwd(const bf16raw*, bf16raw*, uint8_t*, int, int, int, int, int,
                                   int, int, int, int, hipStream_t);
hipError_t launch_maxpool_nhwc_bwd(const bf16raw*, const uint8_t*, bf16raw*, int, int, int, int,
                                   int, int, int, int, int, hipStream_t);
hipError_t launch_gap_nhwc_fwd(const bf16raw*, bf16raw*, int, int, int64_t, hipStream_t);
hipError_t launch_gap_nhwc_bwd(const bf16raw*, bf16raw*, int, int, int64_t, hipStream_t);
hipError_t launch_bn_stats_nhwc(const bf16raw*, float*, float*, float*, float*, float*, float*,
                                int64_t, int, float, float, int, float*, int, hipStream_t);
hipError_t launch_bn_apply_nhwc(const bf16raw*, const bf16raw*, bf16raw*, const float*,
                                const float*, const float*, const float*, int, int64_t, int,
                                hipStream_t);
hipError_t launch_bn_bwd_reduce_nhwc(const bf16raw*, const bf16raw*, const bf16raw*,
                                     const float*, const float*, const float*, const float*,
                                     float*, float*, int64_t, int, int, float*, int,
                                     hipStream_t);
hipError_t launch_bn_bwd_dx_nhwc(const bf16raw*, const bf16raw*, const bf16raw*, const float*,
                                 const float*, const float*, const float*, const float*,
                                 const float*, bf16raw*, int, int64_t, float, int, hipStream_t);
}

#define CHECK_HIP(err)                                                              \
  TORCH_CHECK((err) == hipSuccess, "HIP kernel launch failed: ", hipGetErrorString(err))

static hipStream_t cur_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

static void check_gpu_contig(const at::Tensor& t, c10::ScalarType dt, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == dt, name, " has wrong dtype");
}

// ---------------------------------------------------------------------------

void fused_adam(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v, double lr, double b1,
                double b2, double eps, double wd, double bc1, double bc2, double gscale,
                bool adamw, c10::optional<at::Tensor> step_dev) {
  check_gpu_contig(p, at::kFloat, "p");
  check_gpu_contig(g, at::kFloat, "g");
  const int* sp = nullptr;
  if (step_dev.has_value()) {
    check_gpu_contig(*step_dev, at::kInt, "step_dev");
    sp = step_dev->data_ptr<int>();
  }
  CHECK_HIP(launch_fused_adam(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
                              v.data_ptr<float>(), p.numel(), (float)lr, (float)b1, (float)b2,
                              (float)eps, (float)wd, (float)bc1, (float)bc2, (float)gscale,
                              adamw ? 1 : 0, sp, cur_stream()));
}

void increment_i32(at::Tensor x) {
  check_gpu_contig(x, at::kInt, "x");
  CHECK_HIP(launch_increment_i32(x.data_ptr<int>(), cur_stream()));
}

void fused_sgd(at::Tensor p, at::Tensor g, at::Tensor buf, double lr, double momentum, double wd,
               double dampening, double gscale, bool nesterov, bool first, bool has_momentum) {
  check_gpu_contig(p, at::kFloat, "p");
  CHECK_HIP(launch_fused_sgd(p.data_ptr<float>(), g.data_ptr<float>(), buf.data_ptr<float>(),
                             p.numel(), (float)lr, (float)momentum, (float)wd, (float)dampening,
                             (float)gscale, nesterov ? 1 : 0, first ? 1 : 0, has_momentum ? 1 : 0,
                             cur_stream()));
}

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(y, at::kBFloat16, "y");
  auto dz = at::empty_like(dy);
  CHECK_HIP(launch_relu_bwd((const bf16raw*)dy.data_ptr(), (const bf16raw*)y.data_ptr(),
                            (bf16raw*)dz.data_ptr(), dy.numel(), cur_stream()));
  return dz;
}


// small-N column-sum path: vectorized 8-col groups, partials in scratch
static bool bias_small_path(int64_t M, int N) {
  return N >= 8 && N <= 2048 && (N & 7) == 0 && (256 % (N >> 3)) == 0 && M >= 4096;
}
static int bias_slices(int64_t M, int N) {
  int64_t s = (M * N * 2) / (256 * 1024);
  return (int)(s < 64 ? 64 : (s > 512 ? 512 : s));
}

at::Tensor bias_grad(at::Tensor dz) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  TORCH_CHECK(dz.dim() == 2);
  auto db = at::zeros({dz.size(1)}, dz.options().dtype(at::kFloat));
  int64_t M = dz.size(0);
  int N = (int)dz.size(1);
  at::Tensor scratch;
  float* sp = nullptr;
  int S = 0;
  if (bias_small_path(M, N)) {
    S = bias_slices(M, N);
    scratch = at::empty({(int64_t)S * N}, dz.options().dtype(at::kFloat));
    sp = scratch.data_ptr<float>();
  }
  CHECK_HIP(launch_bias_grad((const bf16raw*)dz.data_ptr(), db.data_ptr<float>(), (int)M, N, sp,
                             S, cur_stream()));
  return db;
}

// ---- implicit-GEMM 3x3 s1 conv (conv_implicit.hip) ----

at::Tensor pad_nhwc(at::Tensor x, int64_t P) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4 && x.size(3) % 8 == 0, "pad_nhwc wants [B,H,W,C], C%8==0");
  int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2), C = (int)x.size(3);
  auto xP = at::empty({B, H + 2 * P, W + 2 * P, C}, x.options());
  CHECK_HIP(launch_pad_nhwc((const bf16raw*)x.data_ptr(), (bf16raw*)xP.data_ptr(), B, H, W, C,
                            (int)P, cur_stream()));
  return xP;
}

at::Tensor s2d_stem(at::Tensor x, int64_t P) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4 && x.size(3) == 3 && x.size(1) % 2 == 0 && x.size(2) % 2 == 0,
              "s2d_stem wants [B, even H, even W, 3]");
  int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2);
  auto out = at::empty({B, H / 2 + 2 * P, W / 2 + 2 * P, 16}, x.options());
  CHECK_HIP(launch_s2d_stem((const bf16raw*)x.data_ptr(), (bf16raw*)out.data_ptr(), B, H, W,
                            (int)P, cur_stream()));
  return out;
}

at::Tensor flip_w2d(at::Tensor w2d, int64_t CI, int64_t KHW) {
  check_gpu_contig(w2d, at::kBFloat16, "w2d");
  TORCH_CHECK(w2d.dim() == 2 && w2d.size(1) == KHW * CI, "w2d must be [CO, KH*KW*CI]");
  int CO = (int)w2d.size(0);
  auto wf = at::empty({CI, KHW * CO}, w2d.options());
  CHECK_HIP(launch_flip_w2d((const bf16raw*)w2d.data_ptr(), (bf16raw*)wf.data_ptr(), CO, (int)CI,
                            (int)KHW, cur_stream()));
  return wf;
}

at::Tensor conv_implicit_fwd(at::Tensor xP, at::Tensor wmat, c10::optional<at::Tensor> bias,
                             int64_t KH, int64_t KW, bool relu, int64_t OHo, int64_t OWo,
                             int64_t SH, int64_t SW) {
  check_gpu_contig(xP, at::kBFloat16, "xP");
  check_gpu_contig(wmat, at::kBFloat16, "wmat");
  int B = (int)xP.size(0), Hp = (int)xP.size(1), Wp = (int)xP.size(2), CI = (int)xP.size(3);
  int OH = (Hp - (int)KH) / (int)SH + 1, OW = (Wp - (int)KW) / (int)SW + 1;
  if (OHo <= 0) OHo = OH;
  if (OWo <= 0) OWo = OW;
  int N = (int)wmat.size(0);
  int Kpad = (int)wmat.size(1);
  TORCH_CHECK(Kpad % 64 == 0 && Kpad >= KH * KW * CI, "wmat k-dim must be K padded to x64");
  TORCH_CHECK(CI % 16 == 0 && N % 8 == 0, "implicit conv wants CI%16==0, N%8==0");
  TORCH_CHECK(OHo <= OH && OWo <= OW, "compact output dims exceed GEMM output");
  auto y = at::empty({(int64_t)B * OHo * OWo, N}, xP.options());
  const float* bp = nullptr;
  if (bias.has_value()) {
    check_gpu_contig(*bias, at::kFloat, "bias");
    bp = bias->data_ptr<float>();
  }
  CHECK_HIP(launch_conv_implicit_fwd((const bf16raw*)xP.data_ptr(),
                                     (const bf16raw*)wmat.data_ptr(), bp, (bf16raw*)y.data_ptr(),
                                     B, Hp, Wp, CI, (int)KH, (int)KW, (int)SH, (int)SW, N, Kpad,
                                     relu ? 1 : 0, (int)OHo, (int)OWo, cur_stream()));
  return y;
}

at::Tensor conv_implicit_wgrad(at::Tensor dz, at::Tensor xP, int64_t KH, int64_t KW,
                               int64_t splitk, bool slab, int64_t OHo, int64_t OWo, int64_t SH,
                               int64_t SW) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  check_gpu_contig(xP, at::kBFloat16, "xP");
  int B = (int)xP.size(0), Hp = (int)xP.size(1), Wp = (int)xP.size(2), CI = (int)xP.size(3);
  int OH = (Hp - (int)KH) / (int)SH + 1, OW = (Wp - (int)KW) / (int)SW + 1;
  if (OHo <= 0) OHo = OH;
  if (OWo <= 0) OWo = OW;
  int CO = (int)dz.size(1);
  TORCH_CHECK(dz.size(0) == (int64_t)B * OHo * OWo, "dz rows must be B*OHo*OWo");
  int64_t N = KH * KW * CI;
  // atomic combine accumulates (needs zeros); the slab reduce writes
  auto dw = slab ? at::empty({CO, N}, xP.options().dtype(at::kFloat))
                 : at::zeros({CO, N}, xP.options().dtype(at::kFloat));
  float* wsp = nullptr;
  at::Tensor ws;
  if (slab) {
    int zs = conv_implicit_wgrad_slices(B * OH * OW, (int)splitk);
    ws = at::empty({zs, CO, N}, dw.options());
    wsp = ws.data_ptr<float>();
  }
  CHECK_HIP(launch_conv_implicit_wgrad((const bf16raw*)dz.data_ptr(),
                                       (const bf16raw*)xP.data_ptr(), dw.data_ptr<float>(), CO,
                                       B, Hp, Wp, CI, (int)KH, (int)KW, (int)SH, (int)SW,
                                       (int)splitk, wsp, (int)OHo, (int)OWo, cur_stream()));
  return dw;
}

at::Tensor cast_f64_f32(at::Tensor src) {
  check_gpu_contig(src, at::kDouble, "src");
  auto dst = at::empty(src.sizes(), src.options().dtype(at::kFloat));
  CHECK_HIP(launch_cast_f64_f32(src.data_ptr<double>(), dst.data_ptr<float>(), src.numel(),
                                cur_stream()));
  return dst;
}

at::Tensor cast_f64_bf16(at::Tensor src) {
  check_gpu_contig(src, at::kDouble, "src");
  auto dst = at::empty(src.sizes(), src.options().dtype(at::kBFloat16));
  CHECK_HIP(launch_cast_f64_bf16(src.data_ptr<double>(), (bf16raw*)dst.data_ptr(), src.numel(),
                                 cur_stream()));
  return dst;
}

at::Tensor cast_f32_bf16(at::Tensor src) {
  check_gpu_contig(src, at::kFloat, "src");
  auto dst = at::empty(src.sizes(), src.options().dtype(at::kBFloat16));
  CHECK_HIP(launch_cast_f32_bf16(src.data_ptr<float>(), (bf16raw*)dst.data_ptr(), src.numel(),
                                 cur_stream()));
  return dst;
}

std::tuple<at::Tensor, at::Tensor> ce_fused(at::Tensor logits, at::Tensor targets) {
  check_gpu_contig(logits, at::kBFloat16, "logits");
  check_gpu_contig(targets, at::kLong, "targets");
  TORCH_CHECK(logits.dim() == 2);
  auto loss = at::zeros({}, logits.options().dtype(at::kFloat));
  auto dlogits = at::empty_like(logits);
  CHECK_HIP(launch_ce_fused((const bf16raw*)logits.data_ptr(), targets.data_ptr<int64_t>(),
                            loss.data_ptr<float>(), (bf16raw*)dlogits.data_ptr(),
                            (int)logits.size(0), (int)logits.size(1), cur_stream()));
  return {loss, dlogits};
}

std::tuple<at::Tensor, at::Tensor> mse_fused(at::Tensor pred, at::Tensor target) {
  check_gpu_contig(pred, at::kBFloat16, "pred");
  check_gpu_contig(target, at::kBFloat16, "target");
  auto loss = at::zeros({}, pred.options().dtype(at::kFloat));
  auto dpred = at::empty_like(pred);
  CHECK_HIP(launch_mse_fused((const bf16raw*)pred.data_ptr(), (const bf16raw*)target.data_ptr(),
                             loss.data_ptr<float>(), (bf16raw*)dpred.data_ptr(), pred.numel(),
                             cur_stream()));
  return {loss, dpred};
}

// ---------------------------------------------------------------------------
// Linear family (see gemm.hip header comment for the stride mappings)
// ---------------------------------------------------------------------------

at::Tensor linear_fwd(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias, bool relu) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(w.is_cuda() && w.is_contiguous(), "w must be contiguous GPU");
  TORCH_CHECK(w.scalar_type() == at::kFloat || w.scalar_type() == at::kBFloat16);
  int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "shape mismatch");
  auto y = at::empty({M, N}, x.options());
  const float* bptr = nullptr;
  int epi = 1;  // EPI_BF16
  if (bias.has_value()) {
    check_gpu_contig(*bias, at::kFloat, "bias");
    bptr = bias->data_ptr<float>();
    epi = relu ? 3 : 2;
  } else if (relu) {
    epi = 4;  // EPI_RELU
  }
  CHECK_HIP(launch_gemm_bf16(x.data_ptr(), w.data_ptr(), w.scalar_type() == at::kFloat ? 1 : 0,
                             nullptr, (bf16raw*)y.data_ptr(), bptr, (int)M, (int)N, (int)K,
                             /*sam*/ K, /*sak*/ 1, /*sbk*/ 1, /*sbn*/ K, epi, 1, nullptr, -1, cur_stream()));
  return y;
}

at::Tensor linear_dgrad(at::Tensor dz, at::Tensor w) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  TORCH_CHECK(w.is_cuda() && w.is_contiguous());
  int64_t M = dz.size(0), N = dz.size(1), K = w.size(1);
  TORCH_CHECK(w.size(0) == N, "shape mismatch");
  auto dx = at::empty({M, K}, dz.options());
  CHECK_HIP(launch_gemm_bf16(dz.data_ptr(), w.data_ptr(), w.scalar_type() == at::kFloat ? 1 : 0,
                             nullptr, (bf16raw*)dx.data_ptr(), nullptr, (int)M, (int)K, (int)N,
                             /*sam*/ N, /*sak*/ 1, /*sbk*/ K, /*sbn*/ 1, 1, 1, nullptr, -1, cur_stream()));
  return dx;
}


// tiny-output wgrad: batched outer product instead of a ~98%-dead MFMA tile
static bool small_wgrad_path(int64_t M, int64_t N, int64_t K) {
  // slot kernel: one lane per (co, 8k) cell group; needs aligned K
  // slot kernel's structural cap is 256 cells (one lane per cell); the wide
  // cell-per-thread variant measured ISSUE-bound (per-CO duplicate col
  // loads) and lost to the dead-tile MFMA path — keep the 2048 limit
  return (K & 7) == 0 && N * K <= 2048 && N <= 64 && K <= 512 && M >= 65536;
}
static int small_wgrad_slices(int64_t M) {
  int64_t s = M / 4096;
  return (int)(s < 256 ? 256 : (s > 2048 ? 2048 : s));
}

at::Tensor linear_wgrad(at::Tensor dz, at::Tensor x, int64_t splitk) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  check_gpu_contig(x, at::kBFloat16, "x");
  int64_t B = dz.size(0), N = dz.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == B, "shape mismatch");
  if (small_wgrad_path(B, N, K)) {
    auto dw = at::zeros({N, K}, x.options().dtype(at::kFloat));
    int S = small_wgrad_slices(B);
    auto scratch = at::empty({(int64_t)S * N * K}, x.options().dtype(at::kFloat));
    CHECK_HIP(launch_small_wgrad((const bf16raw*)dz.data_ptr(), (const bf16raw*)x.data_ptr(),
                                 dw.data_ptr<float>(), scratch.data_ptr<float>(), S, B, (int)N,
                                 (int)K, cur_stream()));
    return dw;
  }
  at::Tensor dw = splitk > 1 ? at::zeros({N, K}, x.options().dtype(at::kFloat))
                             : at::empty({N, K}, x.options().dtype(at::kFloat));
  CHECK_HIP(launch_gemm_bf16(dz.data_ptr(), x.data_ptr(), /*b_is_f32*/ 0, dw.data_ptr<float>(),
                             nullptr, nullptr, (int)N, (int)K, (int)B,
                             /*sam*/ 1, /*sak*/ N, /*sbk*/ K, /*sbn*/ 1, 0, (int)splitk,
                             nullptr, -1, cur_stream()));
  return dw;
}

// Direct-accumulate variants: write into a pre-zeroed (or accumulating) grad
// view — no fresh allocation, no autograd add pass.  Always the atomic
// (accumulate) epilogue so gradient accumulation semantics match autograd.
void linear_wgrad_into(at::Tensor dz, at::Tensor x, at::Tensor dw, int64_t splitk) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  check_gpu_contig(x, at::kBFloat16, "x");
  check_gpu_contig(dw, at::kFloat, "dw");
  int64_t B = dz.size(0), N = dz.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == B && dw.size(0) == N && dw.size(1) == K, "shape mismatch");
  if (small_wgrad_path(B, N, K)) {
    int S = small_wgrad_slices(B);
    auto scratch = at::empty({(int64_t)S * N * K}, x.options().dtype(at::kFloat));
    CHECK_HIP(launch_small_wgrad((const bf16raw*)dz.data_ptr(), (const bf16raw*)x.data_ptr(),
                                 dw.data_ptr<float>(), scratch.data_ptr<float>(), S, B, (int)N,
                                 (int)K, cur_stream()));
    return;
  }
  if (splitk < 1) splitk = 1;
  CHECK_HIP(launch_gemm_bf16(dz.data_ptr(), x.data_ptr(), 0, dw.data_ptr<float>(), nullptr,
                             nullptr, (int)N, (int)K, (int)B, 1, N, K, 1, 0, -(int)splitk,
                             nullptr, -1, cur_stream()));
}

// dW and db in ONE MFMA launch: dW_ext = dz^T @ [x | 1]; the virtual ones
// column's output lands in db via the epilogue (gemm.hip ones_row).
void linear_wgrad_bias_into(at::Tensor dz, at::Tensor x, at::Tensor dw, at::Tensor db,
                            int64_t splitk) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  check_gpu_contig(x, at::kBFloat16, "x");
  check_gpu_contig(dw, at::kFloat, "dw");
  check_gpu_contig(db, at::kFloat, "db");
  int64_t B = dz.size(0), N = dz.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == B && dw.size(0) == N && dw.size(1) == K && db.numel() == N);
  if (splitk < 1) splitk = 1;
  CHECK_HIP(launch_gemm_bf16(dz.data_ptr(), x.data_ptr(), 0, dw.data_ptr<float>(), nullptr,
                             nullptr, (int)N, (int)(K + 1), (int)B, 1, N, K, 1, 0, -(int)splitk,
                             db.data_ptr<float>(), (int)K, cur_stream()));
}

void bias_grad_into(at::Tensor dz, at::Tensor db) {
  check_gpu_contig(dz, at::kBFloat16, "dz");
  check_gpu_contig(db, at::kFloat, "db");
  TORCH_CHECK(dz.dim() == 2 && db.numel() == dz.size(1));
  int64_t M = dz.size(0);
  int N = (int)dz.size(1);
  at::Tensor scratch;
  float* sp = nullptr;
  int S = 0;
  if (bias_small_path(M, N)) {
    S = bias_slices(M, N);
    scratch = at::empty({(int64_t)S * N}, dz.options().dtype(at::kFloat));
    sp = scratch.data_ptr<float>();
  }
  CHECK_HIP(launch_bias_grad((const bf16raw*)dz.data_ptr(), db.data_ptr<float>(), (int)M, N, sp,
                             S, cur_stream()));
}

// ---------------------------------------------------------------------------
// Conv2d lowering + pooling + dropout (conv.hip)
// ---------------------------------------------------------------------------

extern "C" {
hipError_t launch_im2col(const bf16raw*, bf16raw*, int, int, int, int, int, int, int, int, int,
                         int, int, int, hipStream_t);
hipError_t launch_col2im(const bf16raw*, bf16raw*, int, int, int, int, int, int, int, int, int,
                         int, int, int, hipStream_t);
hipError_t launch_maxpool_fwd(const bf16raw*, bf16raw*, uint8_t*, int, int, int, int, int, int,
                              hipStream_t);
hipError_t launch_maxpool_bwd(const bf16raw*, const uint8_t*, bf16raw*, int, int, int, int, int,
                              int, hipStream_t);
hipError_t launch_dropout(const bf16raw*, bf16raw*, int64_t, int64_t, int64_t, float,
                          uint32_t, hipStream_t);
}

at::Tensor im2col(at::Tensor x, int64_t KH, int64_t KW, int64_t sh, int64_t sw, int64_t ph,
                  int64_t pw) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  int64_t B = x.size(0), CI = x.size(1), H = x.size(2), W = x.size(3);
  int64_t HO = (H + 2 * ph - KH) / sh + 1;
  int64_t WO = (W + 2 * pw - KW) / sw + 1;
  auto col = at::empty({B * HO * WO, CI * KH * KW}, x.options());
  CHECK_HIP(launch_im2col((const bf16raw*)x.data_ptr(), (bf16raw*)col.data_ptr(), (int)B, (int)CI,
                          (int)H, (int)W, (int)KH, (int)KW, (int)HO, (int)WO, (int)sh, (int)sw,
                          (int)ph, (int)pw, cur_stream()));
  return col;
}

at::Tensor col2im(at::Tensor dcol, int64_t B, int64_t CI, int64_t H, int64_t W, int64_t KH,
                  int64_t KW, int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  check_gpu_contig(dcol, at::kBFloat16, "dcol");
  int64_t HO = (H + 2 * ph - KH) / sh + 1;
  int64_t WO = (W + 2 * pw - KW) / sw + 1;
  TORCH_CHECK(dcol.size(0) == B * HO * WO && dcol.size(1) == CI * KH * KW);
  auto dx = at::empty({B, CI, H, W}, dcol.options());
  CHECK_HIP(launch_col2im((const bf16raw*)dcol.data_ptr(), (bf16raw*)dx.data_ptr(), (int)B,
                          (int)CI, (int)H, (int)W, (int)KH, (int)KW, (int)HO, (int)WO, (int)sh,
                          (int)sw, (int)ph, (int)pw, cur_stream()));
  return dx;
}

std::tuple<at::Tensor, at::Tensor> maxpool_fwd(at::Tensor x, int64_t ks) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4);
  int64_t B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int64_t HO = H / ks, WO = W / ks;
  auto y = at::empty({B, C, HO, WO}, x.options());
  auto arg = at::empty({B, C, HO, WO}, x.options().dtype(at::kByte));
  CHECK_HIP(launch_maxpool_fwd((const bf16raw*)x.data_ptr(), (bf16raw*)y.data_ptr(),
                               arg.data_ptr<uint8_t>(), (int)(B * C), (int)H, (int)W, (int)ks,
                               (int)HO, (int)WO, cur_stream()));
  return {y, arg};
}

at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor arg, int64_t H, int64_t W, int64_t ks) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  int64_t B = dy.size(0), C = dy.size(1);
  auto dx = at::empty({B, C, H, W}, dy.options());
  // windows are disjoint (stride == kernel): bwd writes every slot, no zero
  CHECK_HIP(launch_maxpool_bwd((const bf16raw*)dy.data_ptr(), arg.data_ptr<uint8_t>(),
                               (bf16raw*)dx.data_ptr(), (int)(B * C), (int)H, (int)W, (int)ks,
                               (int)dy.size(2), (int)dy.size(3), cur_stream()));
  // rows/cols beyond HO*ks / WO*ks (non-divisible inputs) keep garbage: zero
  if (H % ks || W % ks) {
    dx.slice(2, (H / ks) * ks, H).zero_();
    dx.slice(3, (W / ks) * ks, W).zero_();
  }
  return dx;
}

at::Tensor dropout_apply(at::Tensor x, double p, int64_t seed, int64_t units_div,
                         int64_t cmod) {
  check_gpu_contig(x, at::kBFloat16, "x");
  auto y = at::empty_like(x);
  CHECK_HIP(launch_dropout((const bf16raw*)x.data_ptr(), (bf16raw*)y.data_ptr(), x.numel(),
                           units_div, cmod, (float)p, (uint32_t)(seed & 0xffffffff),
                           cur_stream()));
  return y;
}

// generic C = A @ B with optional transposes, for tests/other models
at::Tensor matmul_bf16(at::Tensor a, at::Tensor b, bool trans_a, bool trans_b) {
  check_gpu_contig(a, at::kBFloat16, "a");
  TORCH_CHECK(b.is_cuda() && b.is_contiguous());
  int64_t M = trans_a ? a.size(1) : a.size(0);
  int64_t Ka = trans_a ? a.size(0) : a.size(1);
  int64_t Kb = trans_b ? b.size(1) : b.size(0);
  int64_t N = trans_b ? b.size(0) : b.size(1);
  TORCH_CHECK(Ka == Kb, "inner dims mismatch");
  int64_t sam = trans_a ? 1 : Ka;
  int64_t sak = trans_a ? M : 1;
  int64_t sbk = trans_b ? 1 : N;
  int64_t sbn = trans_b ? Kb : 1;
  auto c = at::empty({M, N}, a.options().dtype(at::kFloat));
  CHECK_HIP(launch_gemm_bf16(a.data_ptr(), b.data_ptr(), b.scalar_type() == at::kFloat ? 1 : 0,
                             c.data_ptr<float>(), nullptr, nullptr, (int)M, (int)N, (int)Ka, sam,
                             sak, sbk, sbn, 0, 1, nullptr, -1, cur_stream()));
  return c;
}

// --------------------------- BatchNorm / ResNet ops ------------------------

static int bn_split(int C) {
  // aim for ~2048 blocks so small-C layers still fill 256 CUs / 8 XCDs
  int s = 2048 / (C > 0 ? C : 1);
  return s < 1 ? 1 : (s > 64 ? 64 : s);
}

// training-mode stats: returns (mean, invstd); updates running stats in-place
// when given.
std::tuple<at::Tensor, at::Tensor> bn_stats(at::Tensor x, c10::optional<at::Tensor> running_mean,
                                            c10::optional<at::Tensor> running_var,
                                            double momentum, double eps) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  int B = (int)x.size(0), C = (int)x.size(1);
  int64_t HW = x.size(2) * x.size(3);
  auto fopt = x.options().dtype(at::kFloat);
  auto sum = at::zeros({C}, fopt), sumsq = at::zeros({C}, fopt);
  auto mean = at::empty({C}, fopt), invstd = at::empty({C}, fopt);
  float* rm = nullptr;
  float* rv = nullptr;
  if (running_mean.has_value()) {
    check_gpu_contig(*running_mean, at::kFloat, "running_mean");
    check_gpu_contig(*running_var, at::kFloat, "running_var");
    rm = running_mean->data_ptr<float>();
    rv = running_var->data_ptr<float>();
  }
  CHECK_HIP(launch_bn_stats((const bf16raw*)x.data_ptr(), sum.data_ptr<float>(),
                            sumsq.data_ptr<float>(), mean.data_ptr<float>(),
                            invstd.data_ptr<float>(), rm, rv, B, C, HW, (float)momentum,
                            (float)eps, bn_split(C), cur_stream()));
  return {mean, invstd};
}

at::Tensor bn_apply(at::Tensor x, c10::optional<at::Tensor> res, at::Tensor mean,
                    at::Tensor invstd, at::Tensor gamma, at::Tensor beta, bool relu) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  const bf16raw* rp = nullptr;
  if (res.has_value()) {
    check_gpu_contig(*res, at::kBFloat16, "res");
    TORCH_CHECK(res->sizes() == x.sizes(), "residual shape mismatch");
    rp = (const bf16raw*)res->data_ptr();
  }
  auto y = at::empty_like(x);
  CHECK_HIP(launch_bn_apply((const bf16raw*)x.data_ptr(), rp, (bf16raw*)y.data_ptr(),
                            mean.data_ptr<float>(), invstd.data_ptr<float>(),
                            gamma.data_ptr<float>(), beta.data_ptr<float>(), (int)x.size(1),
                            x.size(2) * x.size(3), x.numel(), relu ? 1 : 0, cur_stream()));
  return y;
}

// dbeta/dgamma are accumulated IN-PLACE (callers pass pre-zeroed fp32
// buffers — possibly flat-bucket grad views, same convention as
// linear_wgrad_into).
void bn_bwd_reduce(at::Tensor dy, c10::optional<at::Tensor> yrelu, at::Tensor x, at::Tensor mean,
                   at::Tensor invstd, at::Tensor dbeta, at::Tensor dgamma) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(x, at::kBFloat16, "x");
  check_gpu_contig(dbeta, at::kFloat, "dbeta");
  check_gpu_contig(dgamma, at::kFloat, "dgamma");
  const bf16raw* yp = nullptr;
  if (yrelu.has_value()) yp = (const bf16raw*)yrelu->data_ptr();
  int B = (int)x.size(0), C = (int)x.size(1);
  int64_t HW = x.size(2) * x.size(3);
  CHECK_HIP(launch_bn_bwd_reduce((const bf16raw*)dy.data_ptr(), yp, (const bf16raw*)x.data_ptr(),
                                 mean.data_ptr<float>(), invstd.data_ptr<float>(),
                                 dbeta.data_ptr<float>(), dgamma.data_ptr<float>(), B, C, HW,
                                 bn_split(C), cur_stream()));
}

at::Tensor bn_bwd_dx(at::Tensor dy, c10::optional<at::Tensor> yrelu, at::Tensor x,
                     at::Tensor mean, at::Tensor invstd, at::Tensor gamma, at::Tensor dbeta,
                     at::Tensor dgamma, bool train_stats) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(x, at::kBFloat16, "x");
  const bf16raw* yp = nullptr;
  if (yrelu.has_value()) yp = (const bf16raw*)yrelu->data_ptr();
  int C = (int)x.size(1);
  int64_t HW = x.size(2) * x.size(3);
  auto dx = at::empty_like(x);
  float inv_count = 1.0f / (float)(x.size(0) * HW);
  CHECK_HIP(launch_bn_bwd_dx((const bf16raw*)dy.data_ptr(), yp, (const bf16raw*)x.data_ptr(),
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             gamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                             dgamma.data_ptr<float>(), (bf16raw*)dx.data_ptr(), C, HW, x.numel(),
                             inv_count, train_stats ? 1 : 0, cur_stream()));
  return dx;
}

at::Tensor add_relu(at::Tensor a, at::Tensor b) {
  check_gpu_contig(a, at::kBFloat16, "a");
  check_gpu_contig(b, at::kBFloat16, "b");
  TORCH_CHECK(a.sizes() == b.sizes());
  auto out = at::empty_like(a);
  CHECK_HIP(launch_add_relu((const bf16raw*)a.data_ptr(), (const bf16raw*)b.data_ptr(),
                            (bf16raw*)out.data_ptr(), a.numel(), cur_stream()));
  return out;
}

std::tuple<at::Tensor, at::Tensor> maxpool_gen_fwd(at::Tensor x, int64_t ks, int64_t stride,
                                                   int64_t pad) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  TORCH_CHECK(ks <= 15, "kernel too large for uint8 argmax");
  int H = (int)x.size(2), W = (int)x.size(3);
  int HO = (int)((H + 2 * pad - ks) / stride + 1);
  int WO = (int)((W + 2 * pad - ks) / stride + 1);
  int64_t BC = x.size(0) * x.size(1);
  auto y = at::empty({x.size(0), x.size(1), HO, WO}, x.options());
  auto arg = at::empty({x.size(0), x.size(1), HO, WO}, x.options().dtype(at::kByte));
  CHECK_HIP(launch_maxpool_gen_fwd((const bf16raw*)x.data_ptr(), (bf16raw*)y.data_ptr(),
                                   arg.data_ptr<uint8_t>(), BC, H, W, HO, WO, (int)ks,
                                   (int)stride, (int)pad, cur_stream()));
  return {y, arg};
}

at::Tensor maxpool_gen_bwd(at::Tensor dy, at::Tensor arg, int64_t H, int64_t W, int64_t ks,
                           int64_t stride, int64_t pad) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(arg, at::kByte, "arg");
  int HO = (int)dy.size(2), WO = (int)dy.size(3);
  int64_t BC = dy.size(0) * dy.size(1);
  auto dx = at::empty({dy.size(0), dy.size(1), H, W}, dy.options());
  CHECK_HIP(launch_maxpool_gen_bwd((const bf16raw*)dy.data_ptr(), arg.data_ptr<uint8_t>(),
                                   (bf16raw*)dx.data_ptr(), BC, (int)H, (int)W, HO, WO, (int)ks,
                                   (int)stride, (int)pad, cur_stream()));
  return dx;
}

at::Tensor gap_fwd(at::Tensor x) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  auto y = at::empty({x.size(0), x.size(1)}, x.options());
  CHECK_HIP(launch_gap_fwd((const bf16raw*)x.data_ptr(), (bf16raw*)y.data_ptr(),
                           x.size(0) * x.size(1), x.size(2) * x.size(3), cur_stream()));
  return y;
}

at::Tensor gap_bwd(at::Tensor dy, int64_t H, int64_t W) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  auto dx = at::empty({dy.size(0), dy.size(1), H, W}, dy.options());
  CHECK_HIP(launch_gap_bwd((const bf16raw*)dy.data_ptr(), (bf16raw*)dx.data_ptr(), H * W,
                           dx.numel(), cur_stream()));
  return dx;
}

// ------------------------------- NHWC ops ----------------------------------
// x is [B, H, W, C] contiguous; col matrices are [B*HO*WO, KH*KW*CI] with
// (kh, kw, ci) fastest — matching w.permute(0,2,3,1).reshape(CO, K).

at::Tensor im2col_nhwc(at::Tensor x, int64_t KH, int64_t KW, int64_t sh, int64_t sw, int64_t ph,
                       int64_t pw) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be BHWC");
  int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2), CI = (int)x.size(3);
  int HO = (int)((H + 2 * ph - KH) / sh + 1);
  int WO = (int)((W + 2 * pw - KW) / sw + 1);
  // K padded to a multiple of 8: keeps every consumer 16B-aligned (glds GEMM
  // staging, vector wgrad) at the cost of <= 7 zero columns
  int64_t K = KH * KW * CI;
  int64_t Kp = (K + 7) & ~7LL;
  // pad columns are zeroed by the im2col kernels themselves
  auto col = at::empty({(int64_t)B * HO * WO, Kp}, x.options());
  CHECK_HIP(launch_im2col_nhwc((const bf16raw*)x.data_ptr(), (bf16raw*)col.data_ptr(), B, CI, H,
                               W, (int)KH, (int)KW, HO, WO, (int)sh, (int)sw, (int)ph, (int)pw,
                               (int)Kp, cur_stream()));
  return col;
}

at::Tensor col2im_nhwc(at::Tensor dcol, int64_t B, int64_t CI, int64_t H, int64_t W, int64_t KH,
                       int64_t KW, int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  check_gpu_contig(dcol, at::kBFloat16, "dcol");
  int HO = (int)((H + 2 * ph - KH) / sh + 1);
  int WO = (int)((W + 2 * pw - KW) / sw + 1);
  int Kp = (int)dcol.size(1);  // row stride (K padded to x8)
  TORCH_CHECK(Kp >= KH * KW * CI, "dcol narrower than K");
  auto dx = at::empty({B, H, W, CI}, dcol.options());
  CHECK_HIP(launch_col2im_nhwc((const bf16raw*)dcol.data_ptr(), (bf16raw*)dx.data_ptr(), (int)B,
                               (int)CI, (int)H, (int)W, (int)KH, (int)KW, HO, WO, (int)sh,
                               (int)sw, (int)ph, (int)pw, Kp, cur_stream()));
  return dx;
}

std::tuple<at::Tensor, at::Tensor> maxpool_nhwc_fwd(at::Tensor x, int64_t ks, int64_t stride,
                                                    int64_t pad) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be BHWC");
  TORCH_CHECK(ks <= 15, "kernel too large for uint8 argmax");
  int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2), C = (int)x.size(3);
  int HO = (int)((H + 2 * pad - ks) / stride + 1);
  int WO = (int)((W + 2 * pad - ks) / stride + 1);
  auto y = at::empty({B, HO, WO, C}, x.options());
  auto arg = at::empty({B, HO, WO, C}, x.options().dtype(at::kByte));
  CHECK_HIP(launch_maxpool_nhwc_fwd((const bf16raw*)x.data_ptr(), (bf16raw*)y.data_ptr(),
                                    arg.data_ptr<uint8_t>(), B, C, H, W, HO, WO, (int)ks,
                                    (int)stride, (int)pad, cur_stream()));
  return {y, arg};
}

at::Tensor maxpool_nhwc_bwd(at::Tensor dy, at::Tensor arg, int64_t H, int64_t W, int64_t ks,
                            int64_t stride, int64_t pad) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(arg, at::kByte, "arg");
  int B = (int)dy.size(0), HO = (int)dy.size(1), WO = (int)dy.size(2), C = (int)dy.size(3);
  auto dx = at::empty({B, H, W, (int64_t)C}, dy.options());
  CHECK_HIP(launch_maxpool_nhwc_bwd((const bf16raw*)dy.data_ptr(), arg.data_ptr<uint8_t>(),
                                    (bf16raw*)dx.data_ptr(), B, C, (int)H, (int)W, HO, WO,
                                    (int)ks, (int)stride, (int)pad, cur_stream()));
  return dx;
}

at::Tensor gap_nhwc_fwd(at::Tensor x) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be BHWC");
  int B = (int)x.size(0), C = (int)x.size(3);
  auto y = at::empty({B, C}, x.options());
  CHECK_HIP(launch_gap_nhwc_fwd((const bf16raw*)x.data_ptr(), (bf16raw*)y.data_ptr(), B, C,
                                x.size(1) * x.size(2), cur_stream()));
  return y;
}

at::Tensor gap_nhwc_bwd(at::Tensor dy, int64_t H, int64_t W) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  int B = (int)dy.size(0), C = (int)dy.size(1);
  auto dx = at::empty({B, H, W, (int64_t)C}, dy.options());
  CHECK_HIP(launch_gap_nhwc_bwd((const bf16raw*)dy.data_ptr(), (bf16raw*)dx.data_ptr(), B, C,
                                H * W, cur_stream()));
  return dx;
}


// vec reduction path applies when C is a power-of-two in [8, 2048]; S slices
// sized so each covers ~512 KB (few partial rows, no atomics)
static bool bn_vec_path(int C) { return C >= 8 && C <= 2048 && (2048 % C) == 0; }
static int bn_slices(int64_t M, int C) {
  int64_t s = (M * C * 2) / (256 * 1024);  // ~256 KB per slice
  return (int)(s < 64 ? 64 : (s > 512 ? 512 : s));
}

std::tuple<at::Tensor, at::Tensor> bn_stats_nhwc(at::Tensor x,
                                                 c10::optional<at::Tensor> running_mean,
                                                 c10::optional<at::Tensor> running_var,
                                                 double momentum, double eps) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be BHWC");
  int C = (int)x.size(3);
  int64_t M = x.numel() / C;
  auto fopt = x.options().dtype(at::kFloat);
  auto sum = at::zeros({C}, fopt), sumsq = at::zeros({C}, fopt);
  auto mean = at::empty({C}, fopt), invstd = at::empty({C}, fopt);
  float* rm = nullptr;
  float* rv = nullptr;
  if (running_mean.has_value()) {
    check_gpu_contig(*running_mean, at::kFloat, "running_mean");
    check_gpu_contig(*running_var, at::kFloat, "running_var");
    rm = running_mean->data_ptr<float>();
    rv = running_var->data_ptr<float>();
  }
  int tiles = (C + 63) / 64;
  int nsplit = 2048 / (tiles > 0 ? tiles : 1);
  nsplit = nsplit < 1 ? 1 : (nsplit > 256 ? 256 : nsplit);
  at::Tensor scratch;
  float* sp = nullptr;
  int S = 0;
  if (bn_vec_path(C)) {
    S = bn_slices(M, C);
    scratch = at::empty({2LL * S * C}, fopt);
    sp = scratch.data_ptr<float>();
  }
  CHECK_HIP(launch_bn_stats_nhwc((const bf16raw*)x.data_ptr(), sum.data_ptr<float>(),
                                 sumsq.data_ptr<float>(), mean.data_ptr<float>(),
                                 invstd.data_ptr<float>(), rm, rv, M, C, (float)momentum,
                                 (float)eps, nsplit, sp, S, cur_stream()));
  return {mean, invstd};
}

at::Tensor bn_apply_nhwc(at::Tensor x, c10::optional<at::Tensor> res, at::Tensor mean,
                         at::Tensor invstd, at::Tensor gamma, at::Tensor beta, bool relu) {
  check_gpu_contig(x, at::kBFloat16, "x");
  TORCH_CHECK(x.dim() == 4, "x must be BHWC");
  const bf16raw* rp = nullptr;
  if (res.has_value()) {
    check_gpu_contig(*res, at::kBFloat16, "res");
    TORCH_CHECK(res->sizes() == x.sizes(), "residual shape mismatch");
    rp = (const bf16raw*)res->data_ptr();
  }
  auto y = at::empty_like(x);
  CHECK_HIP(launch_bn_apply_nhwc((const bf16raw*)x.data_ptr(), rp, (bf16raw*)y.data_ptr(),
                                 mean.data_ptr<float>(), invstd.data_ptr<float>(),
                                 gamma.data_ptr<float>(), beta.data_ptr<float>(),
                                 (int)x.size(3), x.numel(), relu ? 1 : 0, cur_stream()));
  return y;
}

void bn_bwd_reduce_nhwc(at::Tensor dy, c10::optional<at::Tensor> yrelu, at::Tensor x,
                        at::Tensor mean, at::Tensor invstd, at::Tensor gamma, at::Tensor beta,
                        at::Tensor dbeta, at::Tensor dgamma) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(x, at::kBFloat16, "x");
  check_gpu_contig(dbeta, at::kFloat, "dbeta");
  check_gpu_contig(dgamma, at::kFloat, "dgamma");
  const bf16raw* yp = nullptr;
  if (yrelu.has_value()) yp = (const bf16raw*)yrelu->data_ptr();
  int C = (int)x.size(3);
  int64_t M = x.numel() / C;
  int tiles = (C + 63) / 64;
  int nsplit = 2048 / (tiles > 0 ? tiles : 1);
  nsplit = nsplit < 1 ? 1 : (nsplit > 256 ? 256 : nsplit);
  at::Tensor scratch;
  float* sp = nullptr;
  int S = 0;
  if (bn_vec_path(C)) {
    S = bn_slices(M, C);
    scratch = at::empty({2LL * S * C}, x.options().dtype(at::kFloat));
    sp = scratch.data_ptr<float>();
  }
  CHECK_HIP(launch_bn_bwd_reduce_nhwc((const bf16raw*)dy.data_ptr(), yp,
                                      (const bf16raw*)x.data_ptr(), mean.data_ptr<float>(),
                                      invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                                      beta.data_ptr<float>(), dbeta.data_ptr<float>(),
                                      dgamma.data_ptr<float>(), M, C, nsplit, sp, S,
                                      cur_stream()));
}

at::Tensor bn_bwd_dx_nhwc(at::Tensor dy, c10::optional<at::Tensor> yrelu, at::Tensor x,
                          at::Tensor mean, at::Tensor invstd, at::Tensor gamma, at::Tensor beta,
                          at::Tensor dbeta, at::Tensor dgamma, bool train_stats) {
  check_gpu_contig(dy, at::kBFloat16, "dy");
  check_gpu_contig(x, at::kBFloat16, "x");
  const bf16raw* yp = nullptr;
  if (yrelu.has_value()) yp = (const bf16raw*)yrelu->data_ptr();
  int C = (int)x.size(3);
  auto dx = at::empty_like(x);
  float inv_count = 1.0f / (float)(x.numel() / C);
  CHECK_HIP(launch_bn_bwd_dx_nhwc((const bf16raw*)dy.data_ptr(), yp,
                                  (const bf16raw*)x.data_ptr(), mean.data_ptr<float>(),
                                  invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                                  beta.data_ptr<float>(), dbeta.data_ptr<float>(),
                                  dgamma.data_ptr<float>(), (bf16raw*)dx.data_ptr(), C,
                                  x.numel(), inv_count, train_stats ? 1 : 0, cur_stream()));
  return dx;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_adam", &fused_adam, "fused Adam on a flat bucket");
  m.def("increment_i32", &increment_i32, "device step counter += 1 (graph-capturable)");
  m.def("fused_sgd", &fused_sgd, "fused SGD on a flat bucket");
  m.def("relu_bwd", &relu_bwd, "dz = dy * (y>0)");
  m.def("bias_grad", &bias_grad, "column-sum of dz");
  m.def("pad_nhwc", &pad_nhwc, "zero-pad ring (width P) for implicit conv");
  m.def("s2d_stem", &s2d_stem, "fused space-to-depth + channel pad + ring pad for the 7x7 s2 stem");
  m.def("flip_w2d", &flip_w2d, "dgrad weight transform W[co,k,ci] -> W'[ci,rev(k),co]");
  m.def("conv_implicit_fwd", &conv_implicit_fwd, "implicit-GEMM s1 conv forward / dgrad",
        py::arg("xP"), py::arg("wmat"), py::arg("bias"), py::arg("KH"), py::arg("KW"),
        py::arg("relu"), py::arg("OHo") = -1, py::arg("OWo") = -1,
        py::arg("SH") = 1, py::arg("SW") = 1);
  m.def("conv_implicit_wgrad", &conv_implicit_wgrad, "implicit-GEMM s1 conv weight grad",
        py::arg("dz"), py::arg("xP"), py::arg("KH"), py::arg("KW"), py::arg("splitk"),
        py::arg("slab"), py::arg("OHo") = -1, py::arg("OWo") = -1,
        py::arg("SH") = 1, py::arg("SW") = 1);
  m.def("cast_f64_f32", &cast_f64_f32);
  m.def("cast_f64_bf16", &cast_f64_bf16, "one-pass device Vector pack cast");
  m.def("cast_f32_bf16", &cast_f32_bf16);
  m.def("ce_fused", &ce_fused, "cross-entropy fwd+bwd fused");
  m.def("mse_fused", &mse_fused, "mse fwd+bwd fused");
  m.def("linear_fwd", &linear_fwd, "Y = X W^T (+bias)(+relu), MFMA");
  m.def("linear_dgrad", &linear_dgrad, "dX = dZ W, MFMA");
  m.def("linear_wgrad", &linear_wgrad, "dW = dZ^T X, MFMA split-K");
  m.def("linear_wgrad_into", &linear_wgrad_into, "dW accumulated into grad view");
  m.def("linear_wgrad_bias_into", &linear_wgrad_bias_into, "dW + db in one MFMA launch");
  m.def("bias_grad_into", &bias_grad_into, "db accumulated into grad view");
  m.def("matmul_bf16", &matmul_bf16, "generic bf16 MFMA GEMM");
  m.def("im2col", &im2col, "NCHW -> implicit-GEMM col matrix");
  m.def("col2im", &col2im, "col gradient -> NCHW input gradient");
  m.def("maxpool_fwd", &maxpool_fwd, "max_pool2d (kernel==stride) + argmax");
  m.def("maxpool_bwd", &maxpool_bwd, "max_pool2d backward scatter");
  m.def("dropout_apply", &dropout_apply, "counter-based dropout / dropout2d");
  m.def("bn_stats", &bn_stats, "BatchNorm2d training stats (mean, invstd) + running update");
  m.def("bn_apply", &bn_apply, "BN normalize+affine (+residual)(+relu)");
  m.def("bn_bwd_reduce", &bn_bwd_reduce, "BN backward: dbeta/dgamma accumulated in-place");
  m.def("bn_bwd_dx", &bn_bwd_dx, "BN backward input gradient");
  m.def("add_relu", &add_relu, "out = relu(a+b) (residual join)");
  m.def("maxpool_gen_fwd", &maxpool_gen_fwd, "general max_pool2d (stride/pad) + argmax");
  m.def("maxpool_gen_bwd", &maxpool_gen_bwd, "general max_pool2d backward (gather)");
  m.def("gap_fwd", &gap_fwd, "global average pool [B,C,H,W]->[B,C]");
  m.def("gap_bwd", &gap_bwd, "global average pool backward");
  m.def("im2col_nhwc", &im2col_nhwc, "BHWC -> implicit-GEMM col (vectorized over C)");
  m.def("col2im_nhwc", &col2im_nhwc, "col gradient -> BHWC input gradient");
  m.def("maxpool_nhwc_fwd", &maxpool_nhwc_fwd, "NHWC max_pool2d (stride/pad) + argmax");
  m.def("maxpool_nhwc_bwd", &maxpool_nhwc_bwd, "NHWC max_pool2d backward (gather)");
  m.def("gap_nhwc_fwd", &gap_nhwc_fwd, "NHWC global average pool -> [B,C]");
  m.def("gap_nhwc_bwd", &gap_nhwc_bwd, "NHWC global average pool backward");
  m.def("bn_stats_nhwc", &bn_stats_nhwc, "NHWC BN training stats + running update");
  m.def("bn_apply_nhwc", &bn_apply_nhwc, "NHWC BN normalize+affine (+residual)(+relu)");
  m.def("bn_bwd_reduce_nhwc", &bn_bwd_reduce_nhwc, "NHWC BN backward reduce (in-place)");
  m.def("bn_bwd_dx_nhwc", &bn_bwd_dx_nhwc, "NHWC BN backward input gradient");
}
