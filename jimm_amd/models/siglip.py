"""SigLIP dual-tower model.

Reference: /root/reference/src/jimm/models/siglip.py:15-385. Preserved:
  * vision tower: no pre-norm, patch bias, tanh-gelu, MAP pooling, eps 1e-6
    (siglip.py:60-77); NO visual projection — encode_image returns the MAP
    output (siglip.py:123-133);
  * text tower: NON-causal, eps 1e-6, last-token pooling (needs
    padding="max_length"), BIASED text projection (siglip.py:79-119,145-153);
  * logits = exp(logit_scale) * sim + logit_bias (siglip.py:166-174);
    learnable logit_bias is the SigLIP-specific extra param (siglip.py:121).
"""

from __future__ import annotations

import torch
from torch import nn

from jimm_amd.models.common.text import TextTransformer
from jimm_amd.models.common.vit import VisionTransformerBase


class SigLIP(nn.Module):
    def __init__(
        self,
        image_resolution: int = 256,
        vision_layers: int = 12,
        vision_width: int = 768,
        vision_patch_size: int = 16,
        context_length: int = 64,
        vocab_size: int = 32000,
        transformer_width: int = 768,
        transformer_heads: int = 12,
        transformer_layers: int = 12,
        vision_mlp_dim: int | None = None,
        transformer_mlp_dim: int | None = None,
        *,
        vision_heads: int | None = None,
        layernorm_epsilon: float = 1e-6,
    ) -> None:
        super().__init__()
        vision_heads = vision_heads or max(1, vision_width // 64)  # siglip.py:59
        self.vision_model = VisionTransformerBase(
            img_size=image_resolution,
            patch_size=vision_patch_size,
            hidden_size=vision_width,
            num_layers=vision_layers,
            num_heads=vision_heads,
            mlp_dim=vision_mlp_dim or vision_width * 4,
            use_pre_norm=False,
            use_patch_bias=True,
            hidden_act="gelu_tanh",
            pooling="MAP",
            layernorm_epsilon=layernorm_epsilon,
        )
        self.text_model = TextTransformer(
            vocab_size=vocab_size,
            context_length=context_length,
            hidden_size=transformer_width,
            num_layers=transformer_layers,
            num_heads=transformer_heads,
            mlp_dim=transformer_mlp_dim or transformer_width * 4,
            causal=False,
            pooling="LAST",
            hidden_act="gelu_tanh",
            layernorm_epsilon=layernorm_epsilon,
        )
        self.text_projection = nn.Linear(transformer_width, transformer_width, bias=True)  # biased (siglip.py:111-119)
        self.logit_scale = nn.Parameter(torch.tensor(1.0))
        self.logit_bias = nn.Parameter(torch.tensor(0.0))

    def gradient_checkpointing_enable(self, every_n: int = 1) -> None:
        """Recompute encoder blocks in backward (both towers) — trades ~35%
        step time for ~10x less activation memory (huge-batch training).
        every_n > 1 checkpoints only every n-th block (selective: 1/n the
        recompute cost when the batch leaves some memory headroom)."""
        self.vision_model.encoder.gradient_checkpointing = True
        self.text_model.encoder.gradient_checkpointing = True
        self.vision_model.encoder.checkpoint_every = every_n
        self.text_model.encoder.checkpoint_every = every_n

    def encode_image(self, images: torch.Tensor) -> torch.Tensor:
        return self.vision_model(images)  # no visual projection (siglip.py:123-133)

    def encode_text(self, input_ids: torch.Tensor) -> torch.Tensor:
        from jimm_amd.parallel.tp import row_parallel_linear

        return row_parallel_linear(self.text_model(input_ids), self.text_projection)

    def forward(self, images: torch.Tensor, input_ids: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """Returns (logits_per_image, logits_per_text) — siglip.py:155-174."""
        img = self.encode_image(images)
        txt = self.encode_text(input_ids)
        img = img / img.norm(dim=-1, keepdim=True)
        txt = txt / txt.norm(dim=-1, keepdim=True)
        logits_per_text = self.logit_scale.exp() * txt @ img.t() + self.logit_bias
        return logits_per_text.t(), logits_per_text

    @classmethod
    def from_pretrained(cls, model_name_or_path: str, *, use_pytorch: bool = False, dtype: torch.dtype = torch.float32, device: str | torch.device = "cpu") -> "SigLIP":
        from jimm_amd.interop.siglip_hf import load_siglip

        return load_siglip(cls, model_name_or_path, use_pytorch=use_pytorch, dtype=dtype, device=device)

    def save_pretrained(self, save_dir: str) -> None:
        from jimm_amd.interop.siglip_hf import save_siglip

        save_siglip(self, save_dir)
