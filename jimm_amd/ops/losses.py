"""Training losses (K13) with the distributed variants (C3) and the fused
similarity-head kernels (K12).

The reference ships only the classification CE (examples/vit_training.py:76);
the contrastive / sigmoid losses are implied by the CLIP/SigLIP similarity
heads (models/clip.py:180-188, models/siglip.py:166-174) and the papers.

GPU path: row L2-normalize, softmax-CE over the contrastive logit block and
the SigLIP pairwise sigmoid loss run as HIP kernels (csrc/losses.hip); the
logit GEMMs run through rocBLAS. CPU path is the fp32 torch reference.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn.functional as F

from jimm_amd.ops import _backend
from jimm_amd.parallel.gather import all_gather_with_grad


def softmax_cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean CE with integer labels (examples/vit_training.py:60-78)."""
    if _backend.use_hip(logits):
        return _XentRowsFn.apply(logits.float(), labels)
    return F.cross_entropy(logits.float(), labels)


# ---------------------------------------------------------------------------
# K12 — row L2 normalize
# ---------------------------------------------------------------------------


class _L2NormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        y, rinv = _backend.ext().l2norm_fwd(x)
        ctx.save_for_backward(x, rinv)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, rinv = ctx.saved_tensors
        return _backend.ext().l2norm_bwd(dy, x, rinv)


def l2norm(x: torch.Tensor) -> torch.Tensor:
    if _backend.use_hip(x) and x.dim() == 2:
        return _L2NormFn.apply(x)
    return x / x.norm(dim=-1, keepdim=True)


# ---------------------------------------------------------------------------
# K13 — fused softmax-CE over logit rows (HIP fwd/bwd, mean over rows)
# ---------------------------------------------------------------------------


class _XentRowsFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        logits = logits.contiguous()
        loss, lse = _backend.ext().xent_rows_fwd(logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return loss.mean()

    @staticmethod
    def backward(ctx, g):
        logits, labels, lse = ctx.saved_tensors
        dlogits = _backend.ext().xent_rows_bwd(logits, labels, lse, 1.0 / logits.shape[0])
        return dlogits * g, None


class _SigmoidLossFn(torch.autograd.Function):
    """loss_sum = sum_ij -logsigmoid(z_ij * logits_ij), z = +1 at col diag0+i."""

    @staticmethod
    def forward(ctx, logits, diag0):
        logits = logits.contiguous()
        loss, dlogits = _backend.ext().sigmoid_loss_ew(logits, diag0)
        ctx.save_for_backward(dlogits)
        return loss[0]

    @staticmethod
    def backward(ctx, g):
        (dlogits,) = ctx.saved_tensors
        return dlogits * g, None


# ---------------------------------------------------------------------------
# C3 — distributed contrastive / sigmoid losses
# ---------------------------------------------------------------------------


def clip_contrastive_loss(
    img_emb: torch.Tensor,
    txt_emb: torch.Tensor,
    logit_scale: torch.Tensor,
    *,
    group=None,
    gather: bool = True,
) -> torch.Tensor:
    """Symmetric InfoNCE over the GLOBAL batch.

    Both towers' embeddings are all-gathered (with grad); each rank computes
    its local-rows x global-cols logit blocks; CE is averaged over local
    rows, so the DP gradient average yields the global-batch mean loss.
    """
    img = l2norm(img_emb)
    txt = l2norm(txt_emb)
    scale = logit_scale.exp()
    b_local = img.shape[0]
    do_gather = gather and dist.is_initialized()
    rank = dist.get_rank(group) if do_gather else 0
    img_all = all_gather_with_grad(img, group) if do_gather else img
    txt_all = all_gather_with_grad(txt, group) if do_gather else txt
    labels = torch.arange(b_local, device=img.device) + rank * b_local
    logits_i = scale * img @ txt_all.t()  # (B_local, B_global)
    logits_t = scale * txt @ img_all.t()
    return 0.5 * (softmax_cross_entropy(logits_i, labels) + softmax_cross_entropy(logits_t, labels))


def siglip_sigmoid_loss(
    img_emb: torch.Tensor,
    txt_emb: torch.Tensor,
    logit_scale: torch.Tensor,
    logit_bias: torch.Tensor,
    *,
    group=None,
    gather: bool = True,
    chunk_size: int = 8192,
) -> torch.Tensor:
    """SigLIP pairwise sigmoid loss over the global batch.

    loss = -1/B_global * sum_{i,j} log sigmoid(z_ij * (s*sim_ij + b)),
    z_ij = +1 for matching pairs else -1. Per rank we compute the
    (B_local, B_global) block in column chunks (global batch 32k per
    BASELINE.json config 4 would otherwise materialize 4 GB fp32 logits),
    normalized by B_local so the DP mean reproduces the paper's 1/B_global.
    """
    img = l2norm(img_emb)
    txt = l2norm(txt_emb)
    scale = logit_scale.exp()
    b_local = img.shape[0]
    do_gather = gather and dist.is_initialized()
    rank = dist.get_rank(group) if do_gather else 0
    txt_all = all_gather_with_grad(txt, group) if do_gather else txt
    b_global = txt_all.shape[0]
    use_hip = _backend.use_hip(img)
    diag = torch.arange(b_local, device=img.device)
    total = img.new_zeros((), dtype=torch.float32)
    for start in range(0, b_global, chunk_size):
        cols = txt_all[start : start + chunk_size]
        logits = scale * img @ cols.t() + logit_bias  # (B_local, <=chunk)
        if use_hip:
            total = total + _SigmoidLossFn.apply(logits.float(), rank * b_local - start)
            continue
        z = torch.full_like(logits, -1.0)
        # own positives live at global columns rank*b_local + i
        lo, hi = rank * b_local, rank * b_local + b_local
        if start < hi and lo < start + cols.shape[0]:
            i0 = max(lo, start) - lo
            i1 = min(hi, start + cols.shape[0]) - lo
            z[diag[i0:i1], diag[i0:i1] + lo - start] = 1.0
        total = total - F.logsigmoid(z * logits.float()).sum()
    return total / b_local
