"""GPU linear / patch-embed path: GEMM + fused bias/activation/residual epilogue.

GEMM engine selection (env ``JIMM_AMD_GEMM``):
  * ``hip`` (default) — in-house 8-phase 256-tile MFMA bf16 GEMM
               (csrc/gemm8p.hip) with bias/act/residual/act-bwd epilogues
               fused into the GEMM itself; dW on the split-M TN kernel
               (csrc/gemm_tn8p.hip, fp32-exact); dX on the NT kernel over a
               pre-transposed weight copy.
  * ``blas`` — hipBLASLt/rocBLAS (with the committed MI355X TunableOp
               table) + the separate fused bias+act(+residual) elementwise
               kernel. Kept for A/B and for shapes the in-house kernels
               do not cover (N % 256 != 0 or K % 64 != 0).
  * fp8      — ``jimm_amd.ops.set_fp8(True)``: e4m3 scaled_mm forward and
               plain-dX GEMMs (per-tensor dynamic device-side scales,
               graph-capturable); gradact/dW/LN/losses stay bf16/fp32.
"""

from __future__ import annotations

import os

import torch

from jimm_amd.ops import _backend


# ---------------------------------------------------------------------------
# fp8 forward path (BASELINE config 5): per-tensor dynamic-scaled e4m3 GEMM
# via torch._scaled_mm (hipBLASLt fp8 MFMA — measured 1.4-3.0 PF/s on the
# model shapes, benchmarks/fp8_probe.py, vs ~1.0 PF/s bf16). Backward stays
# bf16 (dX/dW GEMMs on the saved bf16 activations); LN/softmax/losses are
# bf16/fp32 throughout. Enable with jimm_amd.ops.set_fp8(True).
# ---------------------------------------------------------------------------

_FP8_STATE = {"enabled": False}
_FP8_MAX = 448.0  # e4m3 (OCP) max normal


def set_fp8(on: bool = True) -> None:
    _FP8_STATE["enabled"] = bool(on)


def fp8_enabled() -> bool:
    return _FP8_STATE["enabled"]


def _fp8_ok(x2: torch.Tensor, w: torch.Tensor) -> bool:
    # _scaled_mm needs K % 16 == 0 and fp8-capable torch; bf16 inputs only
    return (
        x2.dtype == torch.bfloat16
        and x2.shape[1] % 16 == 0
        and w.shape[0] % 16 == 0
        and hasattr(torch, "_scaled_mm")
    )


def _quant_e4m3(t: torch.Tensor):
    scale = (t.abs().amax().float() / _FP8_MAX).clamp(min=1e-12)
    t8 = (t * scale.reciprocal().to(t.dtype)).to(torch.float8_e4m3fn)
    return t8, scale


def _gemm_nt_fp8(x2: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    x8, sx = _quant_e4m3(x2)
    w8, sw = _quant_e4m3(w)
    return torch._scaled_mm(
        x8, w8.t(), scale_a=sx.view(1, 1), scale_b=sw.view(1, 1), out_dtype=torch.bfloat16
    )


def _gemm_mode() -> str:
    # Default "hip" (round 2): the 8-phase 256-tile MFMA GEMM (csrc/gemm8p)
    # with epilogues (bias/act/residual/act-bwd) fused into the GEMM itself,
    # plus the split-M TN dW kernel (csrc/gemm_tn8p). JIMM_AMD_GEMM=blas
    # switches back to hipBLASLt/rocBLAS + separate elementwise passes.
    return os.environ.get("JIMM_AMD_GEMM", "hip")


def _deterministic() -> bool:
    return os.environ.get("JIMM_AMD_DETERMINISTIC", "0") == "1"


def _dw_ws() -> bool:
    # default workspace+fixed-order-reduce combine (deterministic);
    # JIMM_AMD_DW_WS=0 selects the atomic combine (A/B only)
    return os.environ.get("JIMM_AMD_DW_WS", "1") != "0"


def _dw_gemm(ext, dz: torch.Tensor, x2: torch.Tensor, out_dtype) -> torch.Tensor:
    """dW = dz^T @ x2 — in-house split-M TN kernel (fp32 accumulate; the
    default workspace combine is deterministic, so deterministic mode only
    falls back when JIMM_AMD_DW_WS=0 forces the atomic combine)."""
    if (
        _gemm_mode() == "hip"
        and (_dw_ws() or not _deterministic())
        and dz.dtype == torch.bfloat16
        and ext.gemm_tn8p_supported(dz.shape[0], dz.shape[1], x2.shape[1])
    ):
        return ext.gemm_tn_8p(dz, x2).to(out_dtype)
    return torch.matmul(dz.t(), x2)


def _dw_db_gemm(ext, dz: torch.Tensor, x2: torch.Tensor, out_dtype):
    """dW AND the bias grad in one kernel: db = colsum(dz) is folded out of
    the dW kernel's dz fragments (already in registers) instead of a
    separate full re-read of dz (csrc/gemm_tn8p.hip DBOUT)."""
    if (
        _gemm_mode() == "hip"
        and not _deterministic()
        and dz.dtype == torch.bfloat16
        and ext.gemm_tn8p_supported(dz.shape[0], dz.shape[1], x2.shape[1])
    ):
        dw, db = ext.gemm_tn_8p_db(dz, x2)
        return dw.to(out_dtype), db.to(dz.dtype)
    dw = torch.matmul(dz.t(), x2)
    if dz.is_cuda and dz.shape[-1] % 8 == 0:
        db = _backend.ext().colsum(dz).to(dz.dtype)
    else:
        db = dz.sum(dim=0)
    return dw, db


def _dx_gemm(ext, dz: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """dX = dz @ w — in-house NT kernel on a pre-transposed weight copy.

    (Per-call-quantized fp8 dX was measured NET-NEGATIVE at config 5: the
    w.t().contiguous() + two quantization passes cost more than the fp8
    GEMM saves. The fused-block path instead runs fc1-dX on fp8 with a
    producer-emitted e4m3 operand — see ops/block.py.)"""
    if (
        _gemm_mode() == "hip"
        and dz.dtype == torch.bfloat16
        and ext.gemm8p_supported(dz.shape[0], w.shape[1], w.shape[0])
    ):
        wt = w.t().contiguous()
        y, _ = ext.linear_fwd(dz, wt, None, "", None, False)
        return y
    return torch.matmul(dz, w)


def _gemm_nt(x2: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """z = x2 @ w.T  — (N,K)@(M,K)^T -> (N,M), no epilogue."""
    return torch.matmul(x2, w.t())


class _LinearActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, act, residual):
        in_f, out_f = w.shape[1], w.shape[0]
        x2 = x.contiguous().reshape(-1, in_f)
        res2 = residual.contiguous().reshape(-1, out_f) if residual is not None else None
        ext = _backend.ext()
        if _FP8_STATE["enabled"] and _fp8_ok(x2, w):
            z = _gemm_nt_fp8(x2, w)
            y = ext.bias_act_fwd(z, b, act or "", res2)
        elif _gemm_mode() == "hip" and ext.gemm_supported(x2.shape[0], out_f, in_f, str(x2.dtype)):
            y, z = ext.linear_fwd(x2, w, b, act or "", res2, act is not None)
        elif act is None and res2 is None and b is not None:
            # bias-only: hipBLASLt fuses the bias epilogue into the GEMM —
            # no separate full-tensor bias pass
            y = z = torch.addmm(b, x2, w.t())
        else:
            z = _gemm_nt(x2, w)
            # fused bias+act(+residual): z is updated in place to hold the
            # pre-activation (z+bias); y aliases z when there is no act.
            y = ext.bias_act_fwd(z, b, act or "", res2)
        if act is not None:
            ctx.save_for_backward(x2, w, z)
        else:
            ctx.save_for_backward(x2, w)
        ctx.act = act
        ctx.has_bias = b is not None
        ctx.has_res = residual is not None
        ctx.res_shape = residual.shape if residual is not None else None
        ctx.x_shape = x.shape
        return y.view(*x.shape[:-1], out_f)

    @staticmethod
    def backward(ctx, dy):
        if ctx.act is not None:
            x2, w, z = ctx.saved_tensors
        else:
            x2, w = ctx.saved_tensors
            z = None
        dy2 = dy.contiguous().reshape(-1, dy.shape[-1])
        if ctx.act is not None:
            dz = _backend.ext().act_bwd(dy2, z, ctx.act)
        else:
            dz = dy2
        ext = _backend.ext()
        dx = _dx_gemm(ext, dz, w).view(ctx.x_shape) if ctx.needs_input_grad[0] else None
        if ctx.needs_input_grad[1] and ctx.has_bias and ctx.needs_input_grad[2]:
            dw, db = _dw_db_gemm(ext, dz, x2, w.dtype)
        else:
            dw = _dw_gemm(ext, dz, x2, w.dtype) if ctx.needs_input_grad[1] else None
            if ctx.has_bias and ctx.needs_input_grad[2]:
                if dz.is_cuda and dz.shape[-1] % 8 == 0:
                    db = _backend.ext().colsum(dz).to(dz.dtype)
                else:
                    db = dz.sum(dim=0)
            else:
                db = None
        dres = dy.view(ctx.res_shape) if (ctx.has_res and ctx.needs_input_grad[4]) else None
        return dx, dw, db, None, dres


def linear_act(x, w, b, act, residual):
    return _LinearActFn.apply(x, w, b, act, residual)


class _PatchEmbedFn(torch.autograd.Function):
    """K1: stride=kernel conv == im2col (pure gather) + NT GEMM + bias.

    im2col for kernel==stride is a permute-gather:
      img (B,C,h*P,w*P) -> cols (B*h*w, C*P*P), patch-major rows.
    The HIP kernel does the gather with coalesced writes; the GEMM against
    weight (hidden, C*P*P) reuses the linear path.
    """

    @staticmethod
    def forward(ctx, img, w, b, patch):
        B, C, H, W = img.shape
        h, wn = H // patch, W // patch
        ext = _backend.ext()
        cols = ext.im2col_patch(img.contiguous(), patch)  # (B*h*w, C*P*P)
        w2 = w.reshape(w.shape[0], -1).contiguous()  # (hidden, C*P*P)
        if (
            _gemm_mode() == "hip"
            and cols.dtype == torch.bfloat16
            and ext.gemm8p_supported(cols.shape[0], w2.shape[0], w2.shape[1])
        ):
            y, _ = ext.linear_fwd(cols, w2, b, "", None, False)
        else:
            z = _gemm_nt(cols, w2)
            y = ext.bias_act_fwd(z, b, "", None)
        ctx.save_for_backward(cols, w2)
        ctx.img_shape = img.shape
        ctx.patch = patch
        ctx.has_bias = b is not None
        ctx.w_shape = w.shape
        return y.view(B, h * wn, w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        cols, w2 = ctx.saved_tensors
        patch = ctx.patch
        ext = _backend.ext()
        dy2 = dy.contiguous().reshape(-1, dy.shape[-1])
        dcols = _dx_gemm(ext, dy2, w2)
        dimg = ext.col2im_patch(dcols, list(ctx.img_shape), patch) if ctx.needs_input_grad[0] else None
        dw = _dw_gemm(ext, dy2, cols, w2.dtype).view(ctx.w_shape) if ctx.needs_input_grad[1] else None
        if ctx.has_bias and ctx.needs_input_grad[2]:
            if dy2.shape[-1] % 8 == 0:
                db = _backend.ext().colsum(dy2).to(dy2.dtype)
            else:
                db = dy2.sum(dim=0)
        else:
            db = None
        return dimg, dw, db, None


def patch_embed(img, w, b, patch):
    return _PatchEmbedFn.apply(img, w, b, patch)
