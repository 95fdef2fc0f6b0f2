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
mode and validated against the unsharded oracle at gloo world 2 and 4 for
all three model families (tests/test_distributed_cpu.py).
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


# ---------------------------------------------------------------------------
# Full-model sharding: the reference's (1, n_devices) mesh
# shards EVERY parameter — token embeddings P("model", None)
# (/root/reference/src/jimm/models/clip.py:112-113), the visual/text
# projections (clip.py:89,124-131) and the MAP head. Here:
#   * token embedding: vocab-parallel rows; out-of-range ids contribute a
#     zero vector, partial lookups all-reduced (models/common/text.py);
#   * visual/text projections: row-parallel (shard in-features, matching
#     the reference's P("model", None) on the (in, out) kernel) with the
#     partial products all-reduced;
#   * SigLIP MAP head: q/k/v head-column-parallel, out_proj/fc2 row-parallel
#     (models/common/vit.py MAPHead TP path).
# Deliberately replicated (deviation from the reference, documented):
# LayerNorm scales/biases, position embeddings, the patch-embed conv and the
# ViT classifier head — these are 1-D or tiny (<2 MB total); sharding them
# saves no meaningful memory but inserts an all-gather per use.
# ---------------------------------------------------------------------------


@torch.no_grad()
def shard_vocab_embedding(text_model, group) -> None:
    """Vocab-parallel token embedding: rank keeps rows [start, end)."""
    from torch.nn import Parameter

    ws = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if ws == 1:
        return
    vocab = text_model.token_embedding.weight.shape[0]
    per = (vocab + ws - 1) // ws
    start, end = rank * per, min((rank + 1) * per, vocab)
    text_model.token_embedding.weight = Parameter(
        text_model.token_embedding.weight[start:end].contiguous()
    )
    text_model._tp_vocab = (start, end)
    text_model._tp_group = group if group is not None else dist.group.WORLD


@torch.no_grad()
def shard_row_linear(linear, group) -> None:
    """Row-parallel linear: shard in-features; forward takes the rank's
    input slice and all-reduces the partial product (bias, if any, is added
    after the reduce by the caller)."""
    from torch.nn import Parameter

    ws = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if ws == 1:
        return
    in_f = linear.weight.shape[1]
    assert in_f % ws == 0, f"in_features {in_f} not divisible by tp degree {ws}"
    per = in_f // ws
    linear.weight = Parameter(linear.weight[:, rank * per:(rank + 1) * per].contiguous())
    linear._tp_in = (rank * per, (rank + 1) * per)
    linear._tp_group = group if group is not None else dist.group.WORLD


def row_parallel_linear(x, linear):
    """Apply a (possibly row-parallel-sharded) projection; bias added after
    the reduce so it is counted once."""
    from jimm_amd import ops

    if getattr(linear, "_tp_in", None) is None:
        return ops.linear(x, linear.weight, linear.bias)
    s, e = linear._tp_in
    # copy_to_tp before slicing: the ranks' slice-grads are zero-padded and
    # disjoint, so the backward all-reduce reassembles the full dx exactly
    x = copy_to_tp(x, linear._tp_group)
    part = ops.linear(x[..., s:e], linear.weight)
    out = reduce_from_tp(part, linear._tp_group)
    if linear.bias is not None:
        out = out + linear.bias
    return out


@torch.no_grad()
def shard_map_head(head, group) -> None:
    """Shard the SigLIP MAP pooling head: q/k/v by head blocks (column),
    out_proj/fc2 row-parallel, fc1 column-parallel; probe/LN replicated."""
    from torch.nn import Parameter

    ws = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if ws == 1:
        return
    H = head.hidden_size
    nh, d = head.num_heads, head.head_dim
    assert nh % ws == 0, f"MAP heads {nh} not divisible by tp degree {ws}"
    nh_l = nh // ws
    rows = slice(rank * nh_l * d, (rank + 1) * nh_l * d)
    w, b = head.in_proj_weight, head.in_proj_bias
    head.in_proj_weight = Parameter(
        torch.cat([w[i * H:(i + 1) * H][rows] for i in range(3)], 0).contiguous()
    )
    head.in_proj_bias = Parameter(
        torch.cat([b[i * H:(i + 1) * H][rows] for i in range(3)], 0).contiguous()
    )
    head.out_proj.weight = Parameter(head.out_proj.weight[:, rows].contiguous())
    mlp = head.fc1.weight.shape[0]
    assert mlp % ws == 0
    mrows = slice(rank * (mlp // ws), (rank + 1) * (mlp // ws))
    head.fc1.weight = Parameter(head.fc1.weight[mrows].contiguous())
    head.fc1.bias = Parameter(head.fc1.bias[mrows].contiguous())
    head.fc2.weight = Parameter(head.fc2.weight[:, mrows].contiguous())
    head.num_heads = nh_l
    head._tp_group = group if group is not None else dist.group.WORLD


def shard_vit(model, group=None) -> None:
    """Full-model TP for a VisionTransformer: encoder blocks sharded (and
    the MAP head if present); the classifier head (<3 MB) is replicated."""
    shard_encoder(model.vision.encoder, group)
    if model.vision.map_head is not None:
        shard_map_head(model.vision.map_head, group)


def shard_clip(model, group=None) -> None:
    """Shard every CLIP parameter the reference's mesh mode shards:
    both towers' encoder blocks, the token embedding (vocab-parallel) and
    both projections (row-parallel)."""
    shard_encoder(model.vision_model.encoder, group)
    shard_encoder(model.text_model.encoder, group)
    shard_vocab_embedding(model.text_model, group)
    shard_row_linear(model.visual_projection, group)
    shard_row_linear(model.text_projection, group)


def shard_siglip(model, group=None) -> None:
    """Shard every SigLIP parameter: both encoders, the vocab-parallel token
    embedding, the biased text projection and the MAP pooling head."""
    shard_encoder(model.vision_model.encoder, group)
    shard_encoder(model.text_model.encoder, group)
    shard_vocab_embedding(model.text_model, group)
    shard_row_linear(model.text_projection, group)
    shard_map_head(model.vision_model.map_head, group)
