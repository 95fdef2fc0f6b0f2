"""Text transformer tower shared by CLIP and SigLIP.

Reference semantics:
  * CLIP text (/root/reference/src/jimm/models/clip.py:92-123,148-167):
    token embed + learned pos-emb, CAUSAL transformer, ln_final (eps 1e-5),
    EOT pooling ``x[arange(B), argmax(ids, -1)]`` (EOT=49407 is the max BPE id);
  * SigLIP text (/root/reference/src/jimm/models/siglip.py:79-119,135-153):
    NON-causal transformer, eps 1e-6, last-token pooling ``x[:, -1]``
    (requires padding="max_length").

The projection (bias-free for CLIP, biased for SigLIP) is owned by the model,
not this tower.
"""

from __future__ import annotations

import torch
from torch import nn

from jimm_amd import ops
from jimm_amd.models.common.transformer import Encoder


class TextTransformer(nn.Module):
    def __init__(
        self,
        vocab_size: int,
        context_length: int,
        hidden_size: int,
        num_layers: int,
        num_heads: int,
        mlp_dim: int,
        *,
        causal: bool,
        pooling: str,  # "EOT" | "LAST"
        hidden_act: str = "gelu",
        layernorm_epsilon: float = 1e-5,
        dropout_rate: float = 0.0,
    ) -> None:
        super().__init__()
        if pooling not in ("EOT", "LAST"):
            raise ValueError(f"pooling must be EOT or LAST, got {pooling!r}")
        self.pooling = pooling
        self.eps = layernorm_epsilon
        self.token_embedding = nn.Embedding(vocab_size, hidden_size)
        self.pos_embedding = nn.Parameter(torch.empty(1, context_length, hidden_size))
        nn.init.normal_(self.token_embedding.weight, std=0.02)
        nn.init.normal_(self.pos_embedding, std=0.01)
        self.encoder = Encoder(
            num_layers,
            hidden_size,
            num_heads,
            mlp_dim,
            dropout_rate=dropout_rate,
            hidden_act=hidden_act,
            layernorm_epsilon=layernorm_epsilon,
            causal=causal,
        )
        self.ln_final = nn.LayerNorm(hidden_size, eps=layernorm_epsilon)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        """input_ids (B, L) -> pooled (B, H)."""
        if getattr(self, "_tp_vocab", None) is not None:
            # vocab-parallel embedding (parallel/tp.py): this rank holds
            # token rows [start, end); other ranks' tokens contribute a zero
            # vector; the all-reduce reassembles the full lookup. The
            # position embedding is added AFTER the reduce (it is replicated
            # and must be counted once).
            import torch.nn.functional as F

            from jimm_amd.parallel.tp import reduce_from_tp

            start, end = self._tp_vocab
            local = (input_ids >= start) & (input_ids < end)
            ids_l = (input_ids - start).clamp(0, end - start - 1)
            x = F.embedding(ids_l, self.token_embedding.weight)
            x = x * local.unsqueeze(-1).to(x.dtype)
            x = reduce_from_tp(x, self._tp_group)
            x = x + self.pos_embedding[:, : input_ids.shape[1]].to(x.dtype)
        else:
            x = ops.embed_pos(input_ids, self.token_embedding.weight, self.pos_embedding)  # K10 fused
        x = self.encoder(x)
        x = ops.layer_norm(x, self.ln_final.weight, self.ln_final.bias, self.eps)
        if self.pooling == "EOT":
            # clip.py:164-166 — EOT token has the highest BPE id
            return x[torch.arange(x.shape[0], device=x.device), input_ids.argmax(dim=-1)]
        return x[:, -1]
