"""Autograd wiring for the hand-written gfx950 kernels.

Forward math (bf16 activations, fp32 master weights, fp32 MFMA accumulate):
  y = relu(x @ W^T + b)        — one fused kernel (gemm.hip EPI_BIAS_RELU)
backward:
  dz = dy * (y > 0)            — relu_bwd (elementwise.hip)
  dx = dz @ W                  — MFMA
  dW = dz^T @ x                — MFMA, split-K over batch (atomic fp32)
  db = colsum(dz)              — bias_grad

The CPU fallback keeps identical semantics so the same model code runs in the
no-GPU sandbox; on a GPU the extension is REQUIRED (ops.ext() raises).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from sparktorch_amd import ops


def _choose_splitk(batch: int, n_out: int, k_in: int) -> int:
    """Fill 256 CUs: base grid is ceil(N/128)*ceil(K/128) blocks; split the
    batch reduction until ~1024 blocks (2 blocks/CU x 2 for tail overlap),
    keeping >=512 rows per slice."""
    base = -(-n_out // 128) * (-(-k_in) // 128)
    want = max(1, 2048 // max(1, base))
    max_by_rows = max(1, batch // 512)
    return int(min(want, max_by_rows, 512))


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor], relu: bool):
        # one small cast per step so the GEMM stages a bf16 B by DMA instead
        # of re-reading + converting the fp32 master per M-tile (the cast
        # point is where the stage converted anyway — numerics unchanged)
        wb = ops.ext().cast_f32_bf16(w) if w.dtype == torch.float32 else w
        y = ops.ext().linear_fwd(x, wb, b, relu)
        # wb rides save_for_backward so autograd's saved-tensor versioning
        # covers it (in-place mutation of w between fwd and bwd is detected);
        # only the Parameter reference for _bucket_notify stays an attribute.
        ctx.save_for_backward(x, w, y, wb)
        ctx.relu = relu
        ctx.has_bias = b is not None
        ctx.bias_ref = b
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w, y, wb = ctx.saved_tensors
        ext = ops.ext()
        dy = dy.contiguous()
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dz = ext.relu_bwd(dy, y) if ctx.relu else dy
        dx = ext.linear_dgrad(dz, wb) if ctx.needs_input_grad[0] else None

        # Direct-grad path: when FlatBuckets installed this param, its .grad
        # is a pre-zeroed flat-bucket view — accumulate straight into it (no
        # fresh alloc, no autograd add pass) and notify the bucket so the
        # overlapped all-reduce launches; autograd sees None.
        dw = db = None
        b = ctx.bias_ref
        w_notify = getattr(w, "_bucket_notify", None)
        b_notify = getattr(b, "_bucket_notify", None) if b is not None else None
        sk = _choose_splitk(dz.shape[0], w.shape[0], w.shape[1])
        w_direct = w_notify is not None and w.grad is not None
        b_direct = b_notify is not None and b is not None and b.grad is not None

        # NB: the fused dW+db ones column can push N across a tile boundary
        # (fc2: 257 -> an extra 99.6%-dead N-tile); splitting into wgrad +
        # bias_grad was measured NEUTRAL end-to-end (the separate column-sum
        # pass costs what the dead tile does) — keep the single fused launch.
        if ctx.needs_input_grad[1] and w_direct and ctx.has_bias and b_direct:
            # dW and db in ONE MFMA launch (virtual ones column)
            ext.linear_wgrad_bias_into(dz, x, w.grad, b.grad, sk)
            w_notify()
            b_notify()
        else:
            if ctx.needs_input_grad[1]:
                if w_direct:
                    ext.linear_wgrad_into(dz, x, w.grad, sk)
                    w_notify()
                else:
                    dw = ext.linear_wgrad(dz, x, sk)
            if ctx.has_bias and ctx.needs_input_grad[2]:
                if b_direct:
                    ext.bias_grad_into(dz, b.grad)
                    b_notify()
                else:
                    db = ext.bias_grad(dz)
        return dx, dw, db, None


def hip_linear(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    relu: bool = False,
) -> torch.Tensor:
    """Fused linear (+bias)(+relu).  GPU: hand-written MFMA kernels; CPU:
    identical-semantics torch ops (bf16-sim not applied on CPU)."""
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        if not x.is_contiguous():
            x = x.contiguous()
        if x.dim() > 2:
            # nn.Linear semantics for N-D inputs: apply over the last dim
            lead = x.shape[:-1]
            y = _LinearFn.apply(x.reshape(-1, x.shape[-1]), weight, bias, relu)
            return y.view(*lead, y.shape[-1])
        y = _LinearFn.apply(x, weight, bias, relu)
        return y
    y = F.linear(x, weight, bias)
    return F.relu(y) if relu else y


class _Conv2dFn(torch.autograd.Function):
    """Conv2d as implicit GEMM: im2col -> MFMA linear (+bias)(+relu) ->
    NCHW permute; backward reuses the fused wgrad+bias kernel on the saved
    col matrix and col2im for the input gradient."""

    @staticmethod
    def forward(ctx, x, w, b, stride, padding, relu):
        ext = ops.ext()
        B, CI, H, W = x.shape
        CO, _, KH, KW = w.shape
        sh, sw = stride
        ph, pw = padding
        HO = (H + 2 * ph - KH) // sh + 1
        WO = (W + 2 * pw - KW) // sw + 1
        col = ext.im2col(x, KH, KW, sh, sw, ph, pw)
        w2d = w.reshape(CO, CI * KH * KW).contiguous()
        y2d = ext.linear_fwd(col, w2d, b, relu)  # [B*HO*WO, CO]
        ctx.save_for_backward(col, w2d, y2d)
        ctx.meta = (B, CI, H, W, CO, KH, KW, sh, sw, ph, pw, HO, WO, relu, b is not None)
        ctx.w_ref, ctx.b_ref = w, b
        return y2d.view(B, HO, WO, CO).permute(0, 3, 1, 2).contiguous()

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext()
        col, w2d, y2d = ctx.saved_tensors
        B, CI, H, W, CO, KH, KW, sh, sw, ph, pw, HO, WO, relu, has_bias = ctx.meta
        dy2d = dy.permute(0, 2, 3, 1).reshape(B * HO * WO, CO).contiguous()
        if dy2d.dtype != torch.bfloat16:
            dy2d = dy2d.to(torch.bfloat16)
        dz = ext.relu_bwd(dy2d, y2d) if relu else dy2d

        w, b = ctx.w_ref, ctx.b_ref
        dw = db = None
        sk = _choose_splitk(dz.shape[0], CO, col.shape[1])
        w_notify = getattr(w, "_bucket_notify", None)
        b_notify = getattr(b, "_bucket_notify", None) if b is not None else None
        if ctx.needs_input_grad[1]:
            if w_notify is not None and w.grad is not None and has_bias and b_notify is not None and b.grad is not None:
                ext.linear_wgrad_bias_into(dz, col, w.grad.reshape(CO, -1), b.grad, sk)
                w_notify()
                b_notify()
            else:
                dw = ext.linear_wgrad(dz, col, sk).view(CO, CI, KH, KW)
                if has_bias and ctx.needs_input_grad[2]:
                    db = ext.bias_grad(dz)
        elif has_bias and ctx.needs_input_grad[2]:
            db = ext.bias_grad(dz)

        dx = None
        if ctx.needs_input_grad[0]:
            dcol = ext.linear_dgrad(dz, w2d)
            dx = ext.col2im(dcol, B, CI, H, W, KH, KW, sh, sw, ph, pw)
        return dx, dw, db, None, None, None


def hip_conv2d(x, weight, bias=None, stride=(1, 1), padding=(0, 0), relu=False):
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _Conv2dFn.apply(x.contiguous(), weight, bias, stride, padding, relu)
    y = F.conv2d(x, weight, bias, stride=stride, padding=padding)
    return F.relu(y) if relu else y


class _Conv2dNHWCFn(torch.autograd.Function):
    """Channels-last Conv2d as implicit GEMM.  The profile of the NCHW path
    (profiles/) showed im2col/col2im gather at 54% of the ResNet-18 step;
    in NHWC both kernels move CI-contiguous shortx8 vectors and the GEMM
    output [B*HO*WO, CO] IS the NHWC activation — zero permute kernels.
    Weights stay [CO,CI,KH,KW] (state_dict compatible); the [CO, KH*KW*CI]
    GEMM view is built per forward (small)."""

    @staticmethod
    def forward(ctx, x, w, b, stride, padding, relu):
        ext = ops.ext()
        B, H, W, CI = x.shape
        CO, _, KH, KW = w.shape
        sh, sw = stride
        ph, pw = padding
        HO = (H + 2 * ph - KH) // sh + 1
        WO = (W + 2 * pw - KW) // sw + 1
        col = ext.im2col_nhwc(x, KH, KW, sh, sw, ph, pw)  # [M, Kp] (K padded to x8)
        K = KH * KW * CI
        w2d = w.permute(0, 2, 3, 1).reshape(CO, K)
        if col.shape[1] != K:  # zero-pad weights to the padded K
            w2d = F.pad(w2d, (0, col.shape[1] - K))
        # pre-cast the (small) weight view so BOTH GEMM operands are bf16 and
        # DMA-staged; fp32->bf16 happened at the LDS stage before, so the
        # compute numerics are unchanged
        w2d = w2d.to(torch.bfloat16).contiguous()
        y2d = ext.linear_fwd(col, w2d, b, relu)  # [B*HO*WO, CO] == NHWC
        ctx.save_for_backward(col, w2d, y2d)
        ctx.meta = (B, CI, H, W, CO, KH, KW, sh, sw, ph, pw, HO, WO, relu, b is not None)
        ctx.b_ref = b
        return y2d.view(B, HO, WO, CO)

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext()
        col, w2d, y2d = ctx.saved_tensors
        B, CI, H, W, CO, KH, KW, sh, sw, ph, pw, HO, WO, relu, has_bias = ctx.meta
        dy2d = dy.reshape(B * HO * WO, CO).contiguous()
        if dy2d.dtype != torch.bfloat16:
            dy2d = dy2d.to(torch.bfloat16)
        dz = ext.relu_bwd(dy2d, y2d) if relu else dy2d

        dw = db = None
        sk = _choose_splitk(dz.shape[0], CO, col.shape[1])
        if ctx.needs_input_grad[1]:
            # (kh,kw,ci)-ordered wgrad -> param layout; autograd adds into the
            # flat-bucket grad view, so overlap hooks still fire.  The col
            # matrix is K-padded, so slice the pad columns off first.
            dwp = ext.linear_wgrad(dz, col, sk)
            K = KH * KW * CI
            if dwp.shape[1] != K:
                dwp = dwp[:, :K]
            dw = dwp.reshape(CO, KH, KW, CI).permute(0, 3, 1, 2)
        if has_bias and ctx.needs_input_grad[2]:
            db = ext.bias_grad(dz)

        dx = None
        if ctx.needs_input_grad[0]:
            dcol = ext.linear_dgrad(dz, w2d)
            dx = ext.col2im_nhwc(dcol, B, CI, H, W, KH, KW, sh, sw, ph, pw)
        return dx, dw, db, None, None, None


def _pad64(t2d):
    """Zero-pad a [rows, K] bf16 matrix's K to a multiple of 64 (the implicit
    kernel's LDS chunk) so the weight glds never reads past a row."""
    K = t2d.shape[1]
    Kp = (K + 63) // 64 * 64
    return t2d if Kp == K else F.pad(t2d, (0, Kp - K))


class _ConvImplicitNHWCFn(torch.autograd.Function):
    """Implicit-GEMM conv (conv_implicit.hip): no col matrix at all.
    Forward stages x-patches straight from a zero-ring-padded NHWC tensor
    (any stride); wgrad reads the same padded tensor (L2-served re-reads
    instead of a KH*KW-duplicated col stream); stride-1 dgrad runs the SAME
    forward kernel on pad(dz, KH-1-p) with flipped/transposed weights — no
    dcol, no col2im; stride-2 dgrad is a transposed conv and keeps the
    dcol+col2im scatter (still no im2col).  Also a KH*KW-fold
    activation-memory saving: xP is saved for backward, not the col matrix.
    Eligible: stride in {1,2}, ph == pw <= KH-1, KH == KW, CI % 16 == 0,
    CO % 8 == 0."""

    @staticmethod
    def forward(ctx, x, w, b, stride, padding, relu):
        ext = ops.ext()
        B, H, W, CI = x.shape
        CO, _, KH, KW = w.shape
        p = padding[0]
        s = stride[0]
        K = KH * KW * CI
        w2d = w.permute(0, 2, 3, 1).reshape(CO, K).to(torch.bfloat16).contiguous()
        xP = ext.pad_nhwc(x, p) if p > 0 else x
        y2d = ext.conv_implicit_fwd(xP, _pad64(w2d).contiguous(), b, KH, KW, relu,
                                    -1, -1, s, s)
        OH, OW = (H + 2 * p - KH) // s + 1, (W + 2 * p - KW) // s + 1
        ctx.save_for_backward(xP, w2d, y2d)
        ctx.meta = (B, CI, H, W, CO, KH, KW, p, s, OH, OW, relu, b is not None)
        ctx.b_ref = b
        return y2d.view(B, OH, OW, CO)

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext()
        xP, w2d, y2d = ctx.saved_tensors
        B, CI, H, W, CO, KH, KW, p, s, OH, OW, relu, has_bias = ctx.meta
        dy2d = dy.reshape(B * OH * OW, CO).contiguous()
        if dy2d.dtype != torch.bfloat16:
            dy2d = dy2d.to(torch.bfloat16)
        dz = ext.relu_bwd(dy2d, y2d) if relu else dy2d

        dw = db = None
        if ctx.needs_input_grad[1]:
            # measured sweep (tools/sweep_implicit_wgrad.py on MI355X):
            # best split keeps ~3136 batch rows per K-slice across the
            # resnet shapes (no upper clamp — big batches want more slices);
            # slab combine beats atomics from CO >= 256
            sk = max(16, dz.shape[0] // 3136)
            slab = CO >= 256
            dwp = ext.conv_implicit_wgrad(dz, xP, KH, KW, sk, slab, -1, -1, s, s)
            dw = dwp.reshape(CO, KH, KW, CI).permute(0, 3, 1, 2)
        if has_bias and ctx.needs_input_grad[2]:
            db = ext.bias_grad(dz)

        dx = None
        if ctx.needs_input_grad[0]:
            if s == 1:
                wf = ext.flip_w2d(w2d, CI, KH * KW)
                ring = KH - 1 - p
                dzv = dz.view(B, OH, OW, CO)
                dzP = ext.pad_nhwc(dzv, ring) if ring > 0 else dzv
                dx = ext.conv_implicit_fwd(dzP, _pad64(wf).contiguous(), None, KH, KW, False)
                dx = dx.view(B, H, W, CI)
            else:
                # strided dgrad is a transposed conv — scatter dcol back
                # (col2im only; no im2col anywhere on this path)
                dcol = ext.linear_dgrad(dz, _pad64(w2d).contiguous())
                dx = ext.col2im_nhwc(dcol, B, CI, H, W, KH, KW, s, s, p, p)
        return dx, dw, db, None, None, None


class _ConvStemS2DFn(torch.autograd.Function):
    """ResNet stem (7x7 stride-2 pad-3, CI=3) via space-to-depth: rearrange
    the input into 2x2 pixel blocks (12 channels, zero-padded to 16) so the
    strided 7x7 becomes a dense 4x4 STRIDE-1 conv that the implicit-GEMM
    path runs directly — replacing the issue-bound CI=3 im2col
    (~0.8 ms/step) and the 29%-utilization explicit wgrad.

    Mapping: input row r = 2*oh + kh - 3 = 2*(oh - 2 + kh') + ph with
    kh = 2*kh' + ph - 1, kh' in 0..3, phase ph in 0..1; taps with
    kh outside 0..6 get zero weights.  The symmetric ring-2 pad yields
    113x113 outputs; the valid 112x112 slice is taken (the conv window of
    output oh starts at block row oh-2).  dgrad is not implemented — the
    stem is the first layer (needs_input_grad[0] is False in training).
    """

    _luts = {}  # device -> (fwd_idx[256], fwd_mask[256], inv_idx[147])

    @staticmethod
    def _lut(device):
        key = str(device)
        if key not in _ConvStemS2DFn._luts:
            idx = torch.zeros(256, dtype=torch.long)
            mask = torch.zeros(256)
            inv = torch.zeros(147, dtype=torch.long)
            for khp in range(4):
                for kwp in range(4):
                    for ph in range(2):
                        for pw in range(2):
                            for c in range(4):
                                j = (khp * 4 + kwp) * 16 + (ph * 2 + pw) * 4 + c
                                kh, kw = 2 * khp + ph - 1, 2 * kwp + pw - 1
                                if c < 3 and 0 <= kh <= 6 and 0 <= kw <= 6:
                                    src = c * 49 + kh * 7 + kw
                                    idx[j] = src
                                    mask[j] = 1.0
                                    inv[src] = j
            _ConvStemS2DFn._luts[key] = (
                idx.to(device), mask.to(device), inv.to(device)
            )
        return _ConvStemS2DFn._luts[key]

    @staticmethod
    def forward(ctx, x, w, b, relu):
        assert b is None and not relu, "stem path expects bias folded into BN"
        ext = ops.ext()
        B, H, W, CI = x.shape  # CI == 3
        CO = w.shape[0]
        Hb, Wb = H // 2, W // 2
        # one fused kernel: space-to-depth + channel pad 3->16 + ring pad 2
        xP = ext.s2d_stem(x, 2)  # [B, Hb+4, Wb+4, 16]

        idx, mask, _ = _ConvStemS2DFn._lut(w.device)
        wmat = (w.reshape(CO, 147).index_select(1, idx) * mask).to(torch.bfloat16).contiguous()

        # GEMM output space is (Hb+1)x(Wb+1) (symmetric ring); the kernel
        # compacts to the valid Hb x Wb rows at the C-write — no slice copy
        y = ext.conv_implicit_fwd(xP, wmat, None, 4, 4, False, Hb, Wb)
        ctx.save_for_backward(xP)
        ctx.meta = (B, H, W, CI, CO, Hb, Wb)
        return y.view(B, Hb, Wb, CO)

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext()
        (xP,) = ctx.saved_tensors
        B, H, W, CI, CO, Hb, Wb = ctx.meta
        dw = None
        if ctx.needs_input_grad[1]:
            dz = dy.reshape(B * Hb * Wb, CO)
            if dz.dtype != torch.bfloat16:
                dz = dz.to(torch.bfloat16)
            sk = max(16, min(256, dz.shape[0] // 3136))
            # wgrad reads the compact dz through the same (Hb+1)x(Wb+1)
            # GEMM-space remap (rows beyond Hb/Wb are zero)
            dwp = ext.conv_implicit_wgrad(dz.contiguous(), xP, 4, 4, sk, False, Hb, Wb)
            _, _, inv = _ConvStemS2DFn._lut(xP.device)
            dw = dwp.index_select(1, inv).view(CO, CI, 7, 7)
        if ctx.needs_input_grad[0]:
            raise RuntimeError(
                "stem space-to-depth conv does not produce an input gradient "
                "(it is the first layer); use the explicit conv path instead"
            )
        return None, dw, None, None


def _implicit_eligible(CI, CO, KH, KW, stride, padding):
    return (
        stride[0] == stride[1]
        and stride[0] in (1, 2)
        and KH == KW
        and padding[0] == padding[1]
        and padding[0] <= KH - 1
        and CI % 16 == 0
        and CO % 8 == 0
    )


def hip_conv2d_nhwc(x, weight, bias=None, stride=(1, 1), padding=(0, 0), relu=False):
    """x is [B,H,W,CI] contiguous; returns [B,HO,WO,CO]."""
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        CO, CI_w, KH, KW = weight.shape
        if _implicit_eligible(x.shape[3], CO, KH, KW, stride, padding):
            return _ConvImplicitNHWCFn.apply(
                x.contiguous(), weight, bias, tuple(stride), tuple(padding), relu
            )
        if (
            (KH, KW) == (7, 7)
            and tuple(stride) == (2, 2)
            and tuple(padding) == (3, 3)
            and x.shape[3] == 3
            and CO % 8 == 0
            and x.shape[1] % 2 == 0
            and x.shape[2] % 2 == 0
            and bias is None
            and not relu
            and not x.requires_grad
        ):
            return _ConvStemS2DFn.apply(x.contiguous(), weight, bias, relu)
        return _Conv2dNHWCFn.apply(x.contiguous(), weight, bias, stride, padding, relu)
    y = F.conv2d(x.permute(0, 3, 1, 2), weight, bias, stride=stride, padding=padding)
    y = F.relu(y) if relu else y
    return y.permute(0, 2, 3, 1).contiguous()


class _BatchNorm2dNHWCFn(torch.autograd.Function):
    """NHWC BatchNorm2d: per-channel column reductions (coalesced 64-channel
    tiles), elementwise apply with optional fused ReLU; same direct-grad
    bucket convention as the NCHW variant."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training, momentum, eps, relu):
        ext = ops.ext()
        if training:
            mean, invstd = ext.bn_stats_nhwc(x, running_mean, running_var, momentum, eps)
        else:
            mean = running_mean.contiguous()
            invstd = (running_var + eps).rsqrt().contiguous()
        y = ext.bn_apply_nhwc(x, None, mean, invstd, gamma, beta, relu)
        ctx.save_for_backward(x, y if relu else None, mean, invstd, gamma)
        ctx.relu = relu
        ctx.train_stats = training
        ctx.gamma_ref, ctx.beta_ref = gamma, beta
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext()
        x, yrelu, mean, invstd, gamma = ctx.saved_tensors
        dy = dy.contiguous()
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        g, b = ctx.gamma_ref, ctx.beta_ref
        g_notify = getattr(g, "_bucket_notify", None)
        b_notify = getattr(b, "_bucket_notify", None)
        direct = (
            g_notify is not None and g.grad is not None
            and b_notify is not None and b.grad is not None
        )
        if direct:
            dgamma_buf, dbeta_buf = g.grad, b.grad
        else:
            dgamma_buf = torch.zeros_like(mean)
            dbeta_buf = torch.zeros_like(mean)
        beta = ctx.beta_ref
        ext.bn_bwd_reduce_nhwc(dy, yrelu, x, mean, invstd, gamma, beta, dbeta_buf, dgamma_buf)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext.bn_bwd_dx_nhwc(dy, yrelu, x, mean, invstd, gamma, beta, dbeta_buf,
                                    dgamma_buf, ctx.train_stats)
        if direct:
            g_notify()
            b_notify()
            return dx, None, None, None, None, None, None, None, None
        return dx, dgamma_buf, dbeta_buf, None, None, None, None, None, None


def hip_batch_norm2d_nhwc(x, gamma, beta, running_mean, running_var, training: bool,
                          momentum: float = 0.1, eps: float = 1e-5, relu: bool = False):
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _BatchNorm2dNHWCFn.apply(x.contiguous(), gamma, beta, running_mean, running_var,
                                        training, momentum, eps, relu)
    y = F.batch_norm(x.permute(0, 3, 1, 2), running_mean, running_var, gamma, beta, training,
                     momentum, eps)
    y = F.relu(y) if relu else y
    return y.permute(0, 2, 3, 1).contiguous()


class _MaxPool2dNHWCFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ks, stride, pad):
        y, arg = ops.ext().maxpool_nhwc_fwd(x, ks, stride, pad)
        ctx.save_for_backward(arg)
        ctx.meta = (x.shape[1], x.shape[2], ks, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        H, W, ks, stride, pad = ctx.meta
        dx = ops.ext().maxpool_nhwc_bwd(dy.contiguous().to(torch.bfloat16), arg, H, W, ks,
                                        stride, pad)
        return dx, None, None, None


def hip_max_pool2d_nhwc(x, kernel_size: int, stride: Optional[int] = None, padding: int = 0):
    if stride is None:
        stride = kernel_size
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _MaxPool2dNHWCFn.apply(x.contiguous(), kernel_size, stride, padding)
    y = F.max_pool2d(x.permute(0, 3, 1, 2), kernel_size, stride=stride, padding=padding)
    return y.permute(0, 2, 3, 1).contiguous()


class _ReluFn(torch.autograd.Function):
    """Standalone ReLU on the native path.  Forward reuses the relu_bwd
    kernel as x*(x>0)==relu(x) — no extra kernel needed.  Exists so
    ``relu(maxpool(z))`` can replace ``maxpool(relu(z))`` (equal for the
    monotone max), moving the backward mask onto the pooled (4x smaller for
    a 2x2 pool) tensor instead of the full conv output."""

    @staticmethod
    def forward(ctx, x):
        y = ops.ext().relu_bwd(x, x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        return ops.ext().relu_bwd(dy.contiguous().to(torch.bfloat16), y)


def hip_relu(x):
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _ReluFn.apply(x.contiguous())
    return F.relu(x)


class _GlobalAvgPoolNHWCFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.meta = (x.shape[1], x.shape[2])
        return ops.ext().gap_nhwc_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        H, W = ctx.meta
        return ops.ext().gap_nhwc_bwd(dy.contiguous().to(torch.bfloat16), H, W)


def hip_global_avg_pool_nhwc(x):
    """[B,H,W,C] -> [B,C] channel means."""
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _GlobalAvgPoolNHWCFn.apply(x.contiguous())
    return x.mean(dim=(1, 2))


class _BatchNorm2dFn(torch.autograd.Function):
    """BatchNorm2d on the native kernels: training stats (channel×slice grid
    + atomics, fp32), normalize+affine with optional fused ReLU; backward is
    one reduce kernel (dbeta/dgamma, accumulated into pre-zeroed fp32 buffers
    — flat-bucket grad views on the direct-grad path, like linear_wgrad_into)
    + one elementwise dx kernel.  Replaces the reference's eager CPU
    nn.BatchNorm2d (reference has no BN kernels)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training, momentum, eps, relu):
        ext = ops.ext()
        if training:
            mean, invstd = ext.bn_stats(x, running_mean, running_var, momentum, eps)
        else:
            mean = running_mean.contiguous()
            invstd = (running_var + eps).rsqrt().contiguous()
        y = ext.bn_apply(x, None, mean, invstd, gamma, beta, relu)
        ctx.save_for_backward(x, y if relu else None, mean, invstd, gamma)
        ctx.relu = relu
        ctx.train_stats = training
        ctx.gamma_ref, ctx.beta_ref = gamma, beta
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext()
        x, yrelu, mean, invstd, gamma = ctx.saved_tensors
        dy = dy.contiguous()
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)

        g, b = ctx.gamma_ref, ctx.beta_ref
        g_notify = getattr(g, "_bucket_notify", None)
        b_notify = getattr(b, "_bucket_notify", None)
        direct = (
            g_notify is not None and g.grad is not None
            and b_notify is not None and b.grad is not None
        )
        if direct:
            dgamma_buf, dbeta_buf = g.grad, b.grad
        else:
            dgamma_buf = torch.zeros_like(mean)
            dbeta_buf = torch.zeros_like(mean)
        ext.bn_bwd_reduce(dy, yrelu, x, mean, invstd, dbeta_buf, dgamma_buf)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext.bn_bwd_dx(dy, yrelu, x, mean, invstd, gamma, dbeta_buf, dgamma_buf,
                               ctx.train_stats)
        if direct:
            g_notify()
            b_notify()
            return dx, None, None, None, None, None, None, None, None
        return dx, dgamma_buf, dbeta_buf, None, None, None, None, None, None


def hip_batch_norm2d(x, gamma, beta, running_mean, running_var, training: bool,
                     momentum: float = 0.1, eps: float = 1e-5, relu: bool = False):
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _BatchNorm2dFn.apply(x.contiguous(), gamma, beta, running_mean, running_var,
                                    training, momentum, eps, relu)
    y = F.batch_norm(x, running_mean, running_var, gamma, beta, training, momentum, eps)
    return F.relu(y) if relu else y


class _AddReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        out = ops.ext().add_relu(a, b)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, dy):
        (out,) = ctx.saved_tensors
        dz = ops.ext().relu_bwd(dy.contiguous().to(torch.bfloat16), out)
        return dz, dz


def hip_add_relu(a, b):
    """Residual join: relu(a + b), one fused elementwise kernel."""
    if a.is_cuda:
        if a.dtype != torch.bfloat16:
            a = a.to(torch.bfloat16)
        if b.dtype != torch.bfloat16:
            b = b.to(torch.bfloat16)
        return _AddReluFn.apply(a.contiguous(), b.contiguous())
    return F.relu(a + b)


class _MaxPool2dGenFn(torch.autograd.Function):
    """Overlapping-window max_pool2d (e.g. ResNet stem 3x3/s2/p1); backward
    gathers over covering windows — no atomics."""

    @staticmethod
    def forward(ctx, x, ks, stride, pad):
        y, arg = ops.ext().maxpool_gen_fwd(x, ks, stride, pad)
        ctx.save_for_backward(arg)
        ctx.meta = (x.shape[2], x.shape[3], ks, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        H, W, ks, stride, pad = ctx.meta
        dx = ops.ext().maxpool_gen_bwd(dy.contiguous().to(torch.bfloat16), arg, H, W, ks,
                                       stride, pad)
        return dx, None, None, None


class _GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.meta = (x.shape[2], x.shape[3])
        return ops.ext().gap_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        H, W = ctx.meta
        return ops.ext().gap_bwd(dy.contiguous().to(torch.bfloat16), H, W)


def hip_global_avg_pool(x):
    """[B,C,H,W] -> [B,C] channel means (adaptive_avg_pool2d(x,1).flatten(1))."""
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return _GlobalAvgPoolFn.apply(x.contiguous())
    return F.adaptive_avg_pool2d(x, 1).flatten(1)


class _MaxPool2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ks):
        y, arg = ops.ext().maxpool_fwd(x, ks)
        ctx.save_for_backward(arg)
        ctx.meta = (x.shape[2], x.shape[3], ks)
        return y

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        H, W, ks = ctx.meta
        return ops.ext().maxpool_bwd(dy.contiguous().to(torch.bfloat16), arg, H, W, ks), None


def hip_max_pool2d(x, kernel_size: int, stride: Optional[int] = None, padding: int = 0):
    if stride is None:
        stride = kernel_size
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        x = x.contiguous()
        if stride == kernel_size and padding == 0:
            return _MaxPool2dFn.apply(x, kernel_size)  # disjoint fast path
        return _MaxPool2dGenFn.apply(x, kernel_size, stride, padding)
    return F.max_pool2d(x, kernel_size, stride=stride, padding=padding)


class _DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, seed, units_div, cmod):
        ctx.meta = (p, seed, units_div, cmod)
        return ops.ext().dropout_apply(x, p, seed, units_div, cmod)

    @staticmethod
    def backward(ctx, dy):
        p, seed, units_div, cmod = ctx.meta
        # same (seed, unit) hash -> same mask; scale applies to dy too
        return (ops.ext().dropout_apply(dy.contiguous().to(torch.bfloat16), p, seed, units_div,
                                        cmod), None, None, None, None)


def hip_dropout(x, p: float, training: bool = True, channel_wise: bool = False,
                layout: str = "nchw"):
    if not training or p <= 0.0:
        return x
    if x.is_cuda:
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        cmod = 0
        units_div = 1
        if channel_wise and x.dim() == 4:
            if layout == "nhwc":  # unit (b, c) from [B,H,W,C]: idx/HWC*C + idx%C
                units_div = x.shape[1] * x.shape[2] * x.shape[3]
                cmod = x.shape[3]
            else:
                units_div = x.shape[2] * x.shape[3]
        seed = int(torch.randint(0, 2**31 - 1, (1,)).item())
        return _DropoutFn.apply(x.contiguous(), p, seed, units_div, cmod)
    if channel_wise:
        if layout == "nhwc":
            return F.dropout2d(x.permute(0, 3, 1, 2), p, training).permute(0, 2, 3, 1).contiguous()
        return F.dropout2d(x, p, training)
    return F.dropout(x, p, training)


class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor):
        loss, dlogits = ops.ext().ce_fused(logits, target)
        ctx.save_for_backward(dlogits)
        return loss

    @staticmethod
    def backward(ctx, gout):
        (dlogits,) = ctx.saved_tensors
        return dlogits * gout.to(dlogits.dtype), None


def hip_cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Fused CE fwd+bwd (one kernel computes loss and dlogits)."""
    if logits.is_cuda:
        if logits.dtype != torch.bfloat16:
            logits = logits.to(torch.bfloat16)
        return _CEFn.apply(logits.contiguous(), target.flatten().long().contiguous())
    return F.cross_entropy(logits.float(), target.flatten().long())


class _MSEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pred: torch.Tensor, target: torch.Tensor):
        loss, dpred = ops.ext().mse_fused(pred, target)
        ctx.save_for_backward(dpred)
        return loss

    @staticmethod
    def backward(ctx, gout):
        (dpred,) = ctx.saved_tensors
        return dpred * gout.to(dpred.dtype), None


def hip_mse(pred: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    if pred.is_cuda:
        if pred.dtype != torch.bfloat16:
            pred = pred.to(torch.bfloat16)
        return _MSEFn.apply(pred.contiguous(), target.to(torch.bfloat16).contiguous())
    return F.mse_loss(pred.float(), target.float())
