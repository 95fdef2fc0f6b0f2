"""Vision transformer encoder base + MAP pooling head.

Re-design of /root/reference/src/jimm/common/vit.py for MI355X:
  * patch embedding is an unfold+MFMA-GEMM HIP kernel (K1), not a conv op
    (reference uses nnx.Conv with kernel=stride=patch, vit.py:153-165);
  * images are NCHW (PyTorch/HF convention; the reference's NHWC is a JAX
    idiom and deliberately not copied);
  * CLS-concat + pos-emb add is a fused bandwidth op (K2, vit.py:228-241).

Numerics switches preserved (SURVEY.md §7 gotchas):
  * ``use_pre_norm`` (CLIP's ln_pre) XOR dropout on embeddings (vit.py:238-241)
  * CLS vs MAP pooling (vit.py:245-248); MAP head residual structure
    (vit.py:96-101): out = attn_out + mlp(LN(attn_out)), sliced [:, 0]
  * CLS token init zeros; pos-emb trunc-normal std 0.02 (vit.py:166-171)
  * MAP head intermediate defaults to 4*hidden (vit.py:174-176)
"""

from __future__ import annotations

import torch
from torch import nn

from jimm_amd import ops
from jimm_amd.models.common.transformer import Encoder


class MAPHead(nn.Module):
    """SigLIP Multihead-Attention-Pooling head.

    Reference: /root/reference/src/jimm/common/vit.py:12-101. Parameter
    layout matches torch/HF SigLIP (fused ``in_proj`` q/k/v), which is also
    what the HF checkpoint stores (siglip.py:352-363).
    """

    def __init__(self, hidden_size: int, intermediate_size: int, num_heads: int, layernorm_epsilon: float = 1e-6, hidden_act: str = "gelu_tanh") -> None:
        super().__init__()
        self.hidden_size = hidden_size
        self.num_heads = num_heads
        self.head_dim = hidden_size // num_heads
        self.eps = layernorm_epsilon
        self.probe = nn.Parameter(torch.zeros(1, 1, hidden_size))
        self.in_proj_weight = nn.Parameter(torch.empty(3 * hidden_size, hidden_size))
        self.in_proj_bias = nn.Parameter(torch.zeros(3 * hidden_size))
        self.out_proj = nn.Linear(hidden_size, hidden_size, bias=True)
        self.layernorm = nn.LayerNorm(hidden_size, eps=layernorm_epsilon)
        self.fc1 = nn.Linear(hidden_size, intermediate_size, bias=True)
        self.fc2 = nn.Linear(intermediate_size, hidden_size, bias=True)
        self.act = hidden_act
        nn.init.xavier_uniform_(self.in_proj_weight)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        B, L, H = hidden.shape
        nh, d = self.num_heads, self.head_dim
        wq, wk, wv = self.in_proj_weight.chunk(3, dim=0)
        bq, bk, bv = self.in_proj_bias.chunk(3, dim=0)
        probe = self.probe.expand(B, -1, -1).to(hidden.dtype)
        if getattr(self, "_tp_group", None) is not None:
            # column-parallel q/k/v consume the full (replicated) input; the
            # identity-fwd/all-reduce-bwd makes dhidden/dprobe exact under TP
            from jimm_amd.parallel.tp import copy_to_tp

            hidden = copy_to_tp(hidden, self._tp_group)
            probe = copy_to_tp(probe, self._tp_group)
        q = ops.linear(probe, wq, bq).view(B, 1, nh, d).transpose(1, 2)  # (B, nh, 1, d)
        k = ops.linear(hidden, wk, bk).view(B, L, nh, d).transpose(1, 2)
        v = ops.linear(hidden, wv, bv).view(B, L, nh, d).transpose(1, 2)
        x = ops.attention(q, k, v)  # (B, nh, 1, d) — degenerate flash, Lq=1 (K9)
        x = x.transpose(1, 2).reshape(B, 1, nh * d)
        if getattr(self, "_tp_group", None) is not None:
            # TP path (parallel/tp.py shard_map_head): local heads, partial
            # out-proj / fc2 all-reduced, biases and residual counted once
            from jimm_amd.parallel.tp import copy_to_tp, reduce_from_tp

            g = self._tp_group
            part = ops.linear(x, self.out_proj.weight)
            x = reduce_from_tp(part, g) + self.out_proj.bias
            residual = x
            x = ops.layer_norm(x, self.layernorm.weight, self.layernorm.bias, self.eps)
            x = copy_to_tp(x, g)
            x = ops.linear(x, self.fc1.weight, self.fc1.bias, act=self.act)
            part = ops.linear(x, self.fc2.weight)
            x = reduce_from_tp(part, g) + self.fc2.bias + residual
            return x[:, 0]
        x = ops.linear(x, self.out_proj.weight, self.out_proj.bias)
        residual = x
        x = ops.layer_norm(x, self.layernorm.weight, self.layernorm.bias, self.eps)
        x = ops.linear(x, self.fc1.weight, self.fc1.bias, act=self.act)
        x = ops.linear(x, self.fc2.weight, self.fc2.bias, residual=residual)
        return x[:, 0]


class VisionTransformerBase(nn.Module):
    """ViT encoder shared by ViT / CLIP-vision / SigLIP-vision towers.

    Reference: /root/reference/src/jimm/common/vit.py:104-248.
    """

    def __init__(
        self,
        img_size: int = 224,
        patch_size: int = 16,
        in_channels: int = 3,
        hidden_size: int = 768,
        num_layers: int = 12,
        num_heads: int = 12,
        mlp_dim: int = 3072,
        *,
        dropout_rate: float = 0.0,
        use_pre_norm: bool = False,
        use_patch_bias: bool = True,
        hidden_act: str = "gelu",
        pooling: str = "CLS",  # "CLS" | "MAP" | "NONE"
        layernorm_epsilon: float = 1e-6,
    ) -> None:
        super().__init__()
        if img_size % patch_size != 0:
            raise ValueError(f"img_size {img_size} not divisible by patch_size {patch_size}")
        if pooling not in ("CLS", "MAP", "NONE"):
            raise ValueError(f"pooling must be CLS, MAP or NONE, got {pooling!r}")
        self.img_size = img_size
        self.patch_size = patch_size
        self.hidden_size = hidden_size
        self.pooling = pooling
        self.use_pre_norm = use_pre_norm
        self.eps = layernorm_epsilon
        n_patches = (img_size // patch_size) ** 2

        self.patch_weight = nn.Parameter(torch.empty(hidden_size, in_channels, patch_size, patch_size))
        self.patch_bias = nn.Parameter(torch.zeros(hidden_size)) if use_patch_bias else None
        nn.init.xavier_uniform_(self.patch_weight.view(hidden_size, -1))

        if pooling == "CLS":
            self.cls_token = nn.Parameter(torch.zeros(1, 1, hidden_size))  # zeros init (vit.py:166-168)
            seq_len = n_patches + 1
        else:
            self.cls_token = None
            seq_len = n_patches
        self.pos_embedding = nn.Parameter(torch.empty(1, seq_len, hidden_size))
        nn.init.trunc_normal_(self.pos_embedding, std=0.02)  # (vit.py:169-171)

        if use_pre_norm:
            self.ln_pre = nn.LayerNorm(hidden_size, eps=layernorm_epsilon)
            self.dropout = None
        else:
            self.ln_pre = None
            self.dropout = nn.Dropout(dropout_rate)

        self.encoder = Encoder(
            num_layers,
            hidden_size,
            num_heads,
            mlp_dim,
            dropout_rate=dropout_rate,
            hidden_act=hidden_act,
            layernorm_epsilon=layernorm_epsilon,
        )
        self.ln_post = nn.LayerNorm(hidden_size, eps=layernorm_epsilon)
        if pooling == "MAP":
            # MAP-head MLP width follows the tower's mlp_dim (HF semantics;
            # the reference hardcodes 4*hidden at vit.py:174-176, which only
            # coincides for towers where mlp_dim == 4*hidden)
            self.map_head = MAPHead(hidden_size, mlp_dim, num_heads, layernorm_epsilon, hidden_act=hidden_act)
        else:
            self.map_head = None

    def forward(self, images: torch.Tensor, *, pool: bool = True) -> torch.Tensor:
        x = ops.patch_embed(images, self.patch_weight, self.patch_bias, self.patch_size)  # (B, n_patches, H)
        x = ops.add_cls_pos(x, self.cls_token, self.pos_embedding)
        if self.ln_pre is not None:
            x = ops.layer_norm(x, self.ln_pre.weight, self.ln_pre.bias, self.eps)
        else:
            x = self.dropout(x)
        x = self.encoder(x)
        x = ops.layer_norm(x, self.ln_post.weight, self.ln_post.bias, self.eps)
        if not pool or self.pooling == "NONE":
            return x
        if self.pooling == "CLS":
            return x[:, 0]
        return self.map_head(x)
