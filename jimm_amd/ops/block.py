"""Whole-encoder-block autograd Function (GPU fast path).

The composite EncoderBlock (models/common/transformer.py) is numerically
identical but lets autograd insert two full-tensor grad-accumulation adds
per block (the residual branches: x feeds both LN1 and the attention
residual; a feeds both LN2 and the MLP residual — ~1.1 ms/step for ViT-B
bs256, profiles/r01_NOTES.md). Here the block is one Function and those
adds ride the ln_bwd kernel's fused ``addend`` input for free. The default
path runs every GEMM on the in-house MFMA kernels with the bias / act /
residual / act-backward epilogues fused in; under ``set_fp8(True)`` (and
past the M*H size gate) the forward GEMMs and fc1-dX run e4m3 with
producer-fused quantization and per-block delayed scaling.

Enabled on GPU when dropout == 0 (the composite path remains the reference
and the dropout/CPU path). JIMM_AMD_FUSED_BLOCK=0 disables.
"""

from __future__ import annotations

import os

import torch

from jimm_amd.ops import _backend
from jimm_amd.ops.hip_linear import _dw_db_gemm, _dx_gemm, _gemm_mode


def _hip_gemms(M: int, H: int) -> bool:
    """True when the encoder block's GEMMs run on the in-house MFMA kernels
    (JIMM_AMD_GEMM=hip default; all block shapes have N%256==0, K%64==0)."""
    return _gemm_mode() == "hip" and H % 256 == 0


# minimum rows x hidden for the fp8 block path (tests lower it to force fp8)
_FP8_MIN_MH = 1 << 26


def _fp8_block_state(mod: torch.nn.Module, device: torch.device):
    """Per-block delayed-scaling state, 4 sites: [0] h1->qkv, [1] h2->fc1,
    [2] f->fc2 (forward), [3] dz1->fc1-dX (backward). scale8 is what this
    step's producers divide by; amax collected this step becomes next
    step's scale (scale8[0:3] updated at forward end, [3] at backward
    end). The grad site starts at 1/448 (assume amax~1; adapts in one
    step) so first-step gradients are not flushed to zero."""
    st = getattr(mod, "_fp8_state", None)
    if st is None or st[0].device != device:
        scale = torch.ones(4, device=device, dtype=torch.float32)
        scale[3] = 1.0 / 448.0
        st = (scale, torch.zeros(4, device=device, dtype=torch.float32))
        mod._fp8_state = st
    return st


def _quant_e4m3_t(w: torch.Tensor):
    """Quantize W^T to e4m3 (row-major (K, N) bytes): the column-major b
    operand a scaled_mm dX GEMM (dz @ W) needs is its .t() view."""
    s = (w.abs().amax().float() / 448.0).clamp(min=1e-12)
    w8 = (w * s.reciprocal().to(w.dtype)).to(torch.float8_e4m3fn)  # (N, K) row-major
    return w8.t().contiguous(), s  # (K, N); .t() at the call site is column-major


def _mm_fp8q(y8: torch.Tensor, s: torch.Tensor, w: torch.Tensor,
             bias: torch.Tensor | None = None) -> torch.Tensor:
    """GEMM on a producer-emitted e4m3 activation (uint8 bytes + its scale)
    against a per-call-quantized e4m3 weight; bias fused in the hipBLASLt
    epilogue; bf16 out."""
    from jimm_amd.ops.hip_linear import _quant_e4m3

    w8, sw = _quant_e4m3(w)
    return torch._scaled_mm(
        y8.view(torch.float8_e4m3fn), w8.t(), scale_a=s.view(1, 1),
        scale_b=sw.view(1, 1), bias=bias, out_dtype=torch.bfloat16,
    )


def fused_block_enabled(x: torch.Tensor, dropout_p: float) -> bool:
    return (
        x.is_cuda
        and dropout_p == 0.0
        and x.dtype == torch.bfloat16
        and _backend.use_hip(x)
        and os.environ.get("JIMM_AMD_FUSED_BLOCK", "1") == "1"
    )


class EncoderBlockFn(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx, x, ln1w, ln1b, wqkv, bqkv, wproj, bproj, ln2w, ln2b, w1, b1, w2, b2,
        scale8, amax8,
        num_heads: int, act: str, eps: float, causal: bool, scale: float,
    ):
        ext = _backend.ext()
        B, L, H = x.shape
        x = x.contiguous()
        x2 = x.view(-1, H)
        hip = _hip_gemms(B * L, H)

        # Producer-fused fp8 (BASELINE config 5): LN / bias-act kernels emit
        # the e4m3 copy of their output alongside bf16 (one extra store, no
        # standalone quantization pass) using last step's per-site scale;
        # this step's amax feeds next step's scale (delayed scaling, all
        # device-side — graph-safe). Backward stays bf16 on the saved
        # activations. Sites: h1->qkv, h2->fc1, f->fc2; proj keeps the bf16
        # fused-residual GEMM (its input o comes from attention, not from a
        # producer kernel we control cheaply).
        # fp8 pays only when the tower's GEMMs are big enough to amortize
        # the per-call weight quantization and producer-kernel emission.
        # M*H separates the measured winners from the losers: ViT-L
        # 73856x1024 (+5%) and ViT-B 201728x768 (+8%) win; CLIP-B/32
        # vision 51200x768, CLIP text 78848x512 and SigLIP text 32768x768
        # all lose — those towers stay on the fused bf16 path even under
        # set_fp8(True).
        fp8 = (
            scale8 is not None
            and x.dtype == torch.bfloat16
            and H % 64 == 0
            and H <= 2048
            and B * L * H > _FP8_MIN_MH
            and hasattr(torch, "_scaled_mm")
        )
        if fp8:
            amax8.zero_()
            h1, h1q, mean1, rstd1 = ext.layernorm_fwd_fp8(
                x, ln1w, ln1b, eps, scale8[0:1], amax8[0:1]
            )
            qkv2 = _mm_fp8q(h1q, scale8[0:1], wqkv, bias=bqkv)
        else:
            h1, mean1, rstd1 = ext.layernorm_fwd(x, ln1w, ln1b, eps)
            if hip:
                # in-house MFMA GEMMs with the bias / act / residual epilogues
                # fused into the GEMM kernel itself (csrc/gemm8p.hip)
                qkv2, _ = ext.linear_fwd(h1.view(-1, H), wqkv, bqkv, "", None, False)
            else:
                qkv2 = torch.addmm(bqkv, h1.view(-1, H), wqkv.t())  # (M, 3H) fused bias
        qkv = qkv2.view(B, L, 3, num_heads, H // num_heads)
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        o, lse = ext.attn_fwd(q, k, v, causal, scale)            # (B,nh,L,d), (B,L,nh,d) storage
        o2 = o.transpose(1, 2).reshape(-1, H)                    # free view
        if fp8:
            if hip:
                a, _ = ext.linear_fwd(o2, wproj, bproj, "", x2, False)
            else:
                a = ext.bias_act_fwd(torch.matmul(o2, wproj.t()), bproj, "", x2)
            a3 = a.view(B, L, H)
            h2, h2q, mean2, rstd2 = ext.layernorm_fwd_fp8(
                a3, ln2w, ln2b, eps, scale8[1:2], amax8[1:2]
            )
            z1 = _mm_fp8q(h2q, scale8[1:2], w1, bias=b1)  # z1 = pre-act (bias fused)
            f, f8 = ext.bias_act_fwd_fp8(z1, None, act, scale8[2:3], amax8[2:3])
            y = ext.bias_act_fwd(_mm_fp8q(f8, scale8[2:3], w2, bias=b2), None, "", a)
            # next step's forward scales (delayed); grad sites [3:5] are
            # updated at backward end, after their amax is collected
            scale8[0:3].copy_(torch.clamp(amax8[0:3] / 448.0, min=1e-12))
        elif hip:
            a, _ = ext.linear_fwd(o2, wproj, bproj, "", x2, False)
            a3 = a.view(B, L, H)
            h2, mean2, rstd2 = ext.layernorm_fwd(a3, ln2w, ln2b, eps)
            f, z1 = ext.linear_fwd(h2.view(-1, H), w1, b1, act, None, True)
            y, _ = ext.linear_fwd(f, w2, b2, "", a, False)
        else:
            a2 = torch.matmul(o2, wproj.t())
            a = ext.bias_act_fwd(a2, bproj, "", x2)              # + bias + residual (one pass)
            a3 = a.view(B, L, H)
            h2, mean2, rstd2 = ext.layernorm_fwd(a3, ln2w, ln2b, eps)
            z1 = torch.matmul(h2.view(-1, H), w1.t())
            f = ext.bias_act_fwd(z1, b1, act, None)              # z1 -> pre-activation in place
            y2 = torch.matmul(f, w2.t())
            y = ext.bias_act_fwd(y2, b2, "", a)                  # + bias + residual
        ctx.save_for_backward(x, ln1w, wqkv, wproj, ln2w, w1, w2,
                              h1, qkv, o, lse, a, mean1, rstd1, mean2, rstd2, h2, z1, f)
        ctx.dims = (B, L, H, num_heads)
        ctx.meta = (act, causal, scale)
        ctx.fp8_state = (scale8, amax8) if fp8 else None
        return y.view(B, L, H)

    @staticmethod
    def backward(ctx, dy):
        ext = _backend.ext()
        (x, ln1w, wqkv, wproj, ln2w, w1, w2,
         h1, qkv, o, lse, a, mean1, rstd1, mean2, rstd2, h2, z1, f) = ctx.saved_tensors
        B, L, H, nh = ctx.dims
        act, causal, scale = ctx.meta
        dy2 = dy.contiguous().view(-1, H)
        h1_2 = h1.view(-1, H)
        h2_2 = h2.view(-1, H)
        hip = _hip_gemms(dy2.shape[0], H)
        fp8 = ctx.fp8_state is not None and hip
        if fp8:
            scale8, amax8 = ctx.fp8_state

        # MLP fc2 (+residual into a)
        if fp8:
            # gradact dX with fused e4m3 emission of dz1; the fc1-dX GEMM
            # (dh2 = dz1 @ W1) then runs on the fp8 MFMA pipe
            dz1, dz18 = ext.gemm_nt_8p_gradact_fp8(
                dy2, w2.t().contiguous(), z1, act, scale8[3:4], amax8[3:4]
            )
        elif hip:
            # dX of fc2 with the activation backward fused into the GEMM
            # epilogue: dz1 = (dy @ W2) * act'(z1) — no separate act_bwd pass
            dz1 = ext.gemm_nt_8p_gradact(dy2, w2.t().contiguous(), z1, act)
        else:
            df = torch.matmul(dy2, w2)
            dz1 = ext.act_bwd(df, z1, act)
        dw2, db2 = _dw_db_gemm(ext, dy2, f, w2.dtype)
        if fp8:
            w1t8, sw1 = _quant_e4m3_t(w1)
            dh2 = torch._scaled_mm(
                dz18.view(torch.float8_e4m3fn), w1t8.t(), scale_a=scale8[3:4].view(1, 1),
                scale_b=sw1.view(1, 1), out_dtype=torch.bfloat16,
            )
        else:
            dh2 = _dx_gemm(ext, dz1, w1)
        dw1, db1 = _dw_db_gemm(ext, dz1, h2_2, w1.dtype)
        # LN2 backward with the MLP residual grad (dy) fused into dx
        da3, dln2w, dln2b = ext.layernorm_bwd(
            dh2.view(B, L, H), a.view(B, L, H), ln2w, mean2, rstd2, dy.contiguous()
        )
        da2 = da3.view(-1, H)

        # attention out-projection
        do2 = _dx_gemm(ext, da2, wproj)
        dwproj, dbproj = _dw_db_gemm(ext, da2, o.transpose(1, 2).reshape(-1, H), wproj.dtype)
        do_v = do2.view(B, L, nh, H // nh).permute(0, 2, 1, 3)   # (B,nh,L,d) strided view

        # fused flash attention backward straight into the strided dqkv
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        dqkv = torch.empty_like(qkv)
        ext.attn_bwd_fused(
            q, k, v, o, do_v, lse,
            dqkv[:, :, 0].transpose(1, 2), dqkv[:, :, 1].transpose(1, 2),
            dqkv[:, :, 2].transpose(1, 2), causal, scale,
        )
        dqkv2 = dqkv.view(-1, 3 * H)

        # QKV projection. (qkv-dX stays bf16: quantizing dqkv needs a
        # standalone cast pass — measured a wash vs the fp8 GEMM saving,
        # and it costs dx accuracy. fc1-dX gets its e4m3 operand free from
        # the gradact epilogue, so only that dX runs fp8.)
        dh1 = _dx_gemm(ext, dqkv2, wqkv)
        dwqkv, dbqkv = _dw_db_gemm(ext, dqkv2, h1_2, wqkv.dtype)
        # LN1 backward with the attention residual grad (da) fused into dx
        dx, dln1w, dln1b = ext.layernorm_bwd(
            dh1.view(B, L, H), x, ln1w, mean1, rstd1, da3
        )
        if fp8:
            # next step's backward scale (delayed)
            scale8[3:4].copy_(torch.clamp(amax8[3:4] / 448.0, min=1e-12))
        return (dx, dln1w, dln1b, dwqkv, dbqkv, dwproj, dbproj, dln2w, dln2b,
                dw1, db1, dw2, db2, None, None, None, None, None, None, None)


def encoder_block(x, norm1, qkv, proj, norm2, fc1, fc2, *, num_heads, act, eps, causal, scale):
    from jimm_amd.ops.hip_linear import _FP8_STATE

    scale8 = amax8 = None
    if _FP8_STATE["enabled"] and x.is_cuda and x.dtype == torch.bfloat16:
        scale8, amax8 = _fp8_block_state(norm1, x.device)
    return EncoderBlockFn.apply(
        x, norm1.weight, norm1.bias, qkv.weight, qkv.bias, proj.weight, proj.bias,
        norm2.weight, norm2.bias, fc1.weight, fc1.bias, fc2.weight, fc2.bias,
        scale8, amax8,
        num_heads, act, eps, causal, scale,
    )
