"""All-gather with autograd (C3) — the collective behind the global-batch
contrastive/sigmoid losses.

With the batch sharded across ranks, ``img @ txt.T`` needs one tower's
embeddings from every rank (the reference gets this implicitly from XLA at
/root/reference/src/jimm/models/clip.py:187 under a sharded batch; SURVEY
§2.5 C3). Forward: RCCL all-gather of the local (B_local, H) block over
xGMI. Backward: each rank holds dL_local/d(gathered); the gradient of the
LOCAL block is the SUM over ranks of their slice-r gradients —
a reduce-scatter.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

_last_gather_bytes = 0  # per-call payload, surfaced by Trainer.comm_stats()


class _AllGatherFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        world = dist.get_world_size(group)
        ctx.group = group
        ctx.world = world
        ctx.rank = dist.get_rank(group)
        xc = x.contiguous()
        out = torch.empty(world * x.shape[0], *x.shape[1:], dtype=x.dtype, device=x.device)
        global _last_gather_bytes
        _last_gather_bytes = out.numel() * out.element_size()
        dist.all_gather_into_tensor(out, xc, group=group)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        grad_out = grad_out.contiguous()
        local = torch.empty(
            grad_out.shape[0] // ctx.world, *grad_out.shape[1:], dtype=grad_out.dtype, device=grad_out.device
        )
        if dist.get_backend(ctx.group) == "gloo":
            # gloo has no reduce_scatter_tensor: all-reduce then slice
            dist.all_reduce(grad_out, op=dist.ReduceOp.SUM, group=ctx.group)
            local.copy_(grad_out.chunk(ctx.world, dim=0)[ctx.rank])
        else:
            dist.reduce_scatter_tensor(local, grad_out, op=dist.ReduceOp.SUM, group=ctx.group)
        return local, None


def all_gather_with_grad(x: torch.Tensor, group=None) -> torch.Tensor:
    """Concatenate ``x`` across ranks along dim 0, differentiable."""
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return x
    return _AllGatherFn.apply(x, group)
