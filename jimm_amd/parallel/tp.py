"""Tensor parallelism (C2) — 1-D sharding of the transformer block over
RCCL/xGMI, the explicit equivalent of the reference's declarative weight
sharding (every kernel annotated P(None,"model") etc., activated by
examples/clip_inference.py's mesh (1, n_devices);
/root/reference/src/jimm/common/transformer.py:64-111).

Layout (Megatron-style 1-D):
  * QKV projection: column-parallel by HEAD block — each rank computes its
    num_heads/ws heads end-to-end through the attention kernel;
  * attention out-projection: row-parallel — partial products all-reduced,
    bias added after the reduce;
  * MLP fc1 column-parallel, fc2 row-parallel, same reduce;
  * LayerNorms and biases replicated (their grads are identical across
    ranks for replicated activations).

The two collectives are the classic f/g pair: `copy_to_tp` (identity fwd,
all-reduce bwd) before the column-parallel weights, `reduce_from_tp`
(all-reduce fwd, identity bwd) after the row-parallel weights.

SURVEY notes C2 is optional for parity (models fit easily in 288 GB HBM —
DP-first); it is provided for the reference's pure tensor-sharded inference
mode and validated by the gloo world-2 CPU test.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class _CopyToTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, g):
        g = g.contiguous()
        dist.all_reduce(g, op=dist.ReduceOp.SUM, group=ctx.group)
        return g, None


class _ReduceFromTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, op=dist.ReduceOp.SUM, group=group)
        return x

    @staticmethod
    def backward(ctx, g):
        return g, None


def copy_to_tp(x, group):
    return _CopyToTP.apply(x, group)


def reduce_from_tp(x, group):
    return _ReduceFromTP.apply(x, group)


@torch.no_grad()
def shard_encoder_block(blk, group) -> None:
    """Shard one EncoderBlock in place for tensor parallelism over `group`.

    Each rank keeps num_heads/ws heads of QKV + out-proj and mlp_dim/ws of
    fc1/fc2. Must be applied to a replicated (same-seed or broadcast) block.
    """
    from torch.nn import Parameter

    ws = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if ws == 1:
        return
    H = blk.hidden_size
    nh, d = blk.num_heads, blk.head_dim
    assert nh % ws == 0, f"num_heads {nh} not divisible by tp degree {ws}"
    nh_l = nh // ws
    rows = slice(rank * nh_l * d, (rank + 1) * nh_l * d)

    # qkv weight (3H, H) = [q; k; v] stacked — take the head block of each
    w = blk.qkv.weight
    b = blk.qkv.bias
    parts_w = [w[i * H:(i + 1) * H][rows] for i in range(3)]
    parts_b = [b[i * H:(i + 1) * H][rows] for i in range(3)]
    blk.qkv.weight = Parameter(torch.cat(parts_w, 0).contiguous())
    blk.qkv.bias = Parameter(torch.cat(parts_b, 0).contiguous())
    # out-projection: row-parallel (shard in-features = local head dims)
    blk.proj.weight = Parameter(blk.proj.weight[:, rows].contiguous())
    # MLP
    mlp = blk.fc1.weight.shape[0]
    assert mlp % ws == 0
    mrows = slice(rank * (mlp // ws), (rank + 1) * (mlp // ws))
    blk.fc1.weight = Parameter(blk.fc1.weight[mrows].contiguous())
    blk.fc1.bias = Parameter(blk.fc1.bias[mrows].contiguous())
    blk.fc2.weight = Parameter(blk.fc2.weight[:, mrows].contiguous())

    blk.num_heads = nh_l
    blk._tp_group = group if group is not None else dist.group.WORLD


def shard_encoder(encoder, group) -> None:
    for blk in encoder.layers:
        shard_encoder_block(blk, group)
