"""Training losses (K13) with the distributed variants (C3).

The reference ships only the classification CE (examples/vit_training.py:76);
the contrastive / sigmoid losses are implied by the CLIP/SigLIP similarity
heads (models/clip.py:180-188, models/siglip.py:166-174) and the papers.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn.functional as F

from jimm_amd.parallel.gather import all_gather_with_grad


def softmax_cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean CE with integer labels (examples/vit_training.py:60-78)."""
    return F.cross_entropy(logits.float(), labels)


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
    img = img_emb / img_emb.norm(dim=-1, keepdim=True)
    txt = txt_emb / txt_emb.norm(dim=-1, keepdim=True)
    scale = logit_scale.exp()
    b_local = img.shape[0]
    do_gather = gather and dist.is_initialized()
    rank = dist.get_rank(group) if do_gather else 0
    img_all = all_gather_with_grad(img, group) if do_gather else img
    txt_all = all_gather_with_grad(txt, group) if do_gather else txt
    labels = torch.arange(b_local, device=img.device) + rank * b_local
    logits_i = scale * img @ txt_all.t()  # (B_local, B_global)
    logits_t = scale * txt @ img_all.t()
    return 0.5 * (F.cross_entropy(logits_i.float(), labels) + F.cross_entropy(logits_t.float(), labels))


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
    img = img_emb / img_emb.norm(dim=-1, keepdim=True)
    txt = txt_emb / txt_emb.norm(dim=-1, keepdim=True)
    scale = logit_scale.exp()
    b_local = img.shape[0]
    do_gather = gather and dist.is_initialized()
    rank = dist.get_rank(group) if do_gather else 0
    txt_all = all_gather_with_grad(txt, group) if do_gather else txt
    b_global = txt_all.shape[0]
    diag = torch.arange(b_local, device=img.device)
    total = img.new_zeros(())
    for start in range(0, b_global, chunk_size):
        cols = txt_all[start : start + chunk_size]
        logits = scale * img @ cols.t() + logit_bias  # (B_local, <=chunk)
        z = torch.full_like(logits, -1.0)
        # own positives live at global columns rank*b_local + i
        lo, hi = rank * b_local, rank * b_local + b_local
        if start < hi and lo < start + cols.shape[0]:
            i0 = max(lo, start) - lo
            i1 = min(hi, start + cols.shape[0]) - lo
            z[diag[i0:i1], diag[i0:i1] + lo - start] = 1.0
        total = total + F.logsigmoid(z * logits.float()).sum()
    return -total / b_local
