"""jimm_amd op surface.

Every op dispatches between the hand-written HIP/CDNA4 kernels (GPU path,
``jimm_amd/csrc``) and a pure-PyTorch fp32-friendly reference (CPU path,
which is also the numerics oracle used by tests/).

Kernel inventory implemented here corresponds to SURVEY.md §2.4 (reference
sites cited per op):
  K1/K2  patch_embed (+ cls/pos fusion)  — /root/reference/src/jimm/common/vit.py:153-165,228-241
  K3     layer_norm fwd/bwd              — common/transformer.py:58-66,80-88
  K4..K8 linear (+bias +gelu/quickgelu +residual epilogues)
                                          — common/transformer.py:67-114
  K5     attention (flash-style, causal opt) — common/transformer.py:130, models/clip.py:62
  K14    fused Adam                       — examples/vit_training.py:202-203 (optax.adam)
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from jimm_amd.ops import _backend

# ---------------------------------------------------------------------------
# activations
# ---------------------------------------------------------------------------


def quickgelu(x: torch.Tensor) -> torch.Tensor:
    """OpenAI CLIP activation: x * sigmoid(1.702 x).

    Reference: /root/reference/src/jimm/common/transformer.py:12-19.
    """
    return x * torch.sigmoid(1.702 * x)


def _act(x: torch.Tensor, act: str | None) -> torch.Tensor:
    if act is None:
        return x
    if act == "gelu":
        return F.gelu(x)  # exact erf gelu, matches jax.nn.gelu(approximate=False)? see note
    if act == "gelu_tanh":
        return F.gelu(x, approximate="tanh")
    if act == "quickgelu":
        return quickgelu(x)
    raise ValueError(f"unknown activation {act!r}")


# ---------------------------------------------------------------------------
# K3 — LayerNorm
# ---------------------------------------------------------------------------


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x = x.contiguous()
        y, mean, rstd = _backend.ext().layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = _backend.ext().layernorm_bwd(dy.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float) -> torch.Tensor:
    if _backend.use_hip(x):
        return _LayerNormFn.apply(x, weight, bias, eps)
    # CPU reference: fp32 math for low-precision inputs, float64 preserved
    ref_dtype = torch.float64 if x.dtype == torch.float64 else torch.float32
    y = F.layer_norm(x.to(ref_dtype), (x.shape[-1],), weight.to(ref_dtype), bias.to(ref_dtype), eps)
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# K5 — scaled-dot-product attention (flash-style on GPU)
# q, k, v: (B, H, Lq, D) / (B, H, Lk, D); returns (B, H, Lq, D)
# ---------------------------------------------------------------------------


def _strided_ok(t: torch.Tensor) -> bool:
    return t.stride(3) == 1


def _attn_bwd_composite(ext, q, k, v, o, do, lse, causal, scale):
    """Recompute-P composite via rocBLAS batched GEMMs (fallback path,
    JIMM_AMD_ATTN_BWD=composite); materializes (B,H,Lq,Lk)."""
    q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
    o, do = o.contiguous(), do.contiguous()
    s = torch.matmul(q, k.transpose(-1, -2))  # (B,H,Lq,Lk) bf16
    ext.attn_bwd_p(s, lse, causal, scale)     # in place: s -> P (bf16)
    dcoef = ext.attn_d(do, o)                 # (rows,) fp32 rowsum(dO*O)
    dp = torch.matmul(do, v.transpose(-1, -2))
    ext.attn_ds(dp, s, dcoef, scale)          # in place: dp -> dS (bf16)
    dq = torch.matmul(dp, k)
    dk = torch.matmul(dp.transpose(-1, -2), q)
    dv = torch.matmul(s.transpose(-1, -2), do)
    return dq, dk, dv


def _use_fused_bwd() -> bool:
    import os

    return os.environ.get("JIMM_AMD_ATTN_BWD", "fused") == "fused"


class _AttentionFn(torch.autograd.Function):
    """Forward: fused flash HIP kernel (csrc/attention.hip), saves lse.

    Backward: fused flash backward (csrc/attention_bwd_fused.hip) — two MFMA
    kernels (dK/dV over kv tiles, dQ over q tiles) recomputing P from lse,
    reading/writing strided views with no permute copies. The round-1 profile
    measured the old recompute-P rocBLAS composite at ~15 ms/step for ViT-B
    (profiles/r01_NOTES.md), ~4 ms of it pure `.contiguous()` copies.
    """

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        # kernel consumes strided (B,H,L,64) views directly (e.g. slices of
        # the fused qkv projection) — no permute copies on the forward path
        o, lse = _backend.ext().attn_fwd(q, k, v, causal, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ext = _backend.ext()
        if _use_fused_bwd() and all(_strided_ok(t) for t in (q, k, v, o, do)):
            dq, dk, dv = torch.empty_like(q), torch.empty_like(k), torch.empty_like(v)
            ext.attn_bwd_fused(q, k, v, o, do, lse, dq, dk, dv, ctx.causal, ctx.scale)
        else:
            dq, dk, dv = _attn_bwd_composite(ext, q, k, v, o, do, lse, ctx.causal, ctx.scale)
        return dq, dk, dv, None, None


class _AttentionQKVFn(torch.autograd.Function):
    """Self-attention taking the fused QKV projection output directly.

    Input qkv: (B, L, 3, H, D) (a view of the (B,L,3H*D) GEMM output).
    Avoids autograd's per-slice backward (which materializes three
    full-size zero tensors + scatter + add per layer): backward assembles
    dqkv with three strided copies into one buffer.
    Output: (B, H, L, D) view of (B,L,H,D) storage (free reshape to (B,L,HD)).
    """

    @staticmethod
    def forward(ctx, qkv, causal, scale):
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        o, lse = _backend.ext().attn_fwd(q, k, v, causal, scale)
        ctx.save_for_backward(qkv, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, o, lse = ctx.saved_tensors
        scale = ctx.scale
        ext = _backend.ext()
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        dqkv = torch.empty_like(qkv)
        if _use_fused_bwd() and all(_strided_ok(t) for t in (o, do)):
            # fused kernels write dq/dk/dv straight into the strided dqkv
            # buffer — zero assembly copies
            ext.attn_bwd_fused(
                q, k, v, o, do, lse,
                dqkv[:, :, 0].transpose(1, 2), dqkv[:, :, 1].transpose(1, 2),
                dqkv[:, :, 2].transpose(1, 2), ctx.causal, scale,
            )
        else:
            dq, dk, dv = _attn_bwd_composite(ext, q, k, v, o, do, lse, ctx.causal, scale)
            dqkv[:, :, 0].transpose(1, 2).copy_(dq)
            dqkv[:, :, 1].transpose(1, 2).copy_(dk)
            dqkv[:, :, 2].transpose(1, 2).copy_(dv)
        return dqkv, None, None


def attention_qkv(qkv: torch.Tensor, *, causal: bool = False, scale: float | None = None) -> torch.Tensor:
    """qkv (B, L, 3, H, D) -> (B, H, L, D). GPU fast path for self-attention.

    The flash kernels are bf16 with head_dim in {64, 72, 80, 96, 128}
    (DP-padded MFMA templates); other dtypes/head dims
    run the composite torch math (on GPU or CPU alike)."""
    if scale is None:
        scale = 1.0 / math.sqrt(qkv.shape[-1])
    if _backend.use_hip(qkv) and qkv.dtype == torch.bfloat16 and qkv.shape[-1] in (64, 72, 80, 96, 128):
        return _AttentionQKVFn.apply(qkv, causal, scale)
    q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
    return attention(q, k, v, causal=causal, scale=scale)


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    *,
    causal: bool = False,
    scale: float | None = None,
) -> torch.Tensor:
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _backend.use_hip(q) and q.dtype == torch.bfloat16 and q.shape[-1] in (64, 72, 80, 96, 128):
        return _AttentionFn.apply(q, k, v, causal, scale)
    # composite reference (fp32 math; float64 preserved for gradcheck)
    ref_dtype = torch.float64 if q.dtype == torch.float64 else torch.float32
    qf, kf, vf = q.to(ref_dtype), k.to(ref_dtype), v.to(ref_dtype)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        Lq, Lk = s.shape[-2], s.shape[-1]
        mask = torch.ones(Lq, Lk, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, vf).to(q.dtype)


# ---------------------------------------------------------------------------
# K4/K6/K7/K8 — linear with fused epilogues (bias, activation, residual add)
# x: (..., in_f); weight: (out_f, in_f) torch convention; returns (..., out_f)
# ---------------------------------------------------------------------------


def linear(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor | None = None,
    *,
    act: str | None = None,
    residual: torch.Tensor | None = None,
) -> torch.Tensor:
    """act(x @ weight.T + bias) [+ residual] with the epilogue fused on GPU."""
    if _backend.use_hip(x):
        return _linear_hip(x, weight, bias, act, residual)
    y = F.linear(x, weight, bias)
    y = _act(y, act)
    if residual is not None:
        y = y + residual
    return y


def _linear_hip(x, weight, bias, act, residual):
    # Defined in terms of composable autograd pieces so backward is correct:
    # the GEMM runs through torch.matmul (rocBLAS) or the in-house MFMA GEMM
    # depending on JIMM_AMD_GEMM; bias+act(+residual) run in one fused HIP
    # elementwise kernel with a fused backward.
    import jimm_amd.ops.hip_linear as hip_linear

    return hip_linear.linear_act(x, weight, bias, act, residual)


# ---------------------------------------------------------------------------
# K1/K2 — patch embedding (+CLS concat + pos-emb add fusion)
# img: (B, C, H, W); weight: (hidden, C, P, P); returns (B, n_patches, hidden)
# ---------------------------------------------------------------------------


def patch_embed(
    img: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor | None,
    patch: int,
) -> torch.Tensor:
    if _backend.use_hip(img):
        import jimm_amd.ops.hip_linear as hip_linear

        return hip_linear.patch_embed(img, weight, bias, patch)
    y = F.conv2d(img, weight, bias, stride=patch)  # (B, hidden, h, w)
    return y.flatten(2).transpose(1, 2)  # (B, n_patches, hidden)


class _ClsPosFn(torch.autograd.Function):
    """K2 fused on GPU: one kernel does CLS-tile + concat + pos add."""

    @staticmethod
    def forward(ctx, x, cls_token, pos_emb):
        ctx.has_cls = cls_token is not None
        cls = cls_token.to(x.dtype).view(1, 1, -1) if cls_token is not None else None
        Lout = x.shape[1] + (1 if ctx.has_cls else 0)
        pos = pos_emb[0, :Lout].to(x.dtype)
        return _backend.ext().cls_pos_fwd(x.contiguous(), cls, pos)

    @staticmethod
    def backward(ctx, dy):
        if ctx.has_cls:
            dx = dy[:, 1:].contiguous()
            dcls = dy[:, :1].sum(dim=0, keepdim=True)
        else:
            dx, dcls = dy, None
        dpos = dy.sum(dim=0, keepdim=True)
        return dx, dcls, dpos


def add_cls_pos(
    x: torch.Tensor,
    cls_token: torch.Tensor | None,
    pos_emb: torch.Tensor,
) -> torch.Tensor:
    """[CLS concat] + position-embedding add (K2): pure bandwidth.

    On GPU a single HIP kernel (csrc/elementwise.hip cls_pos_kernel) tiles the
    CLS token, concatenates, and adds the position embedding in one pass; the
    CPU path is the eager cat+add oracle.
    Reference semantics: /root/reference/src/jimm/common/vit.py:232-241.
    """
    if _backend.use_hip(x) and x.shape[-1] % 8 == 0:
        return _ClsPosFn.apply(x, cls_token, pos_emb)
    if cls_token is not None:
        cls = cls_token.expand(x.shape[0], -1, -1).to(x.dtype)
        x = torch.cat([cls, x], dim=1)
    return x + pos_emb[:, : x.shape[1]].to(x.dtype)


class _EmbedPosFn(torch.autograd.Function):
    """K10 fused on GPU: token-embedding gather + pos-emb add in one kernel.

    Reference semantics: /root/reference/src/jimm/models/clip.py:159-160,
    siglip.py:146-147.
    """

    @staticmethod
    def forward(ctx, ids, emb, pos_emb):
        ctx.save_for_backward(ids)
        ctx.vocab = emb.shape[0]
        pos = pos_emb[0, : ids.shape[-1]].to(emb.dtype)
        return _backend.ext().embed_pos_fwd(ids, emb, pos)

    @staticmethod
    def backward(ctx, dy):
        (ids,) = ctx.saved_tensors
        demb = None
        if ctx.needs_input_grad[1]:
            demb = torch.zeros(
                ctx.vocab, dy.shape[-1], dtype=dy.dtype, device=dy.device
            )
            demb.index_add_(0, ids.reshape(-1), dy.reshape(-1, dy.shape[-1]))
        dpos = dy.sum(dim=0).unsqueeze(0) if ctx.needs_input_grad[2] else None
        return None, demb, dpos


def embed_pos(ids: torch.Tensor, emb: torch.Tensor, pos_emb: torch.Tensor) -> torch.Tensor:
    """Token-embedding lookup + position-embedding add (K10), fused on GPU."""
    if _backend.use_hip(emb) and emb.shape[-1] % 8 == 0:
        return _EmbedPosFn.apply(ids, emb, pos_emb)
    return torch.nn.functional.embedding(ids, emb) + pos_emb[:, : ids.shape[-1]].to(emb.dtype)


__all__ = [
    "quickgelu",
    "layer_norm",
    "attention",
    "attention_qkv",
    "linear",
    "patch_embed",
    "add_cls_pos",
    "embed_pos",
]
