"""CLIP dual-tower model.

Reference: /root/reference/src/jimm/models/clip.py:15-416. Preserved:
  * vision tower: pre-norm (ln_pre), NO patch bias, quickgelu, CLS pooling,
    eps 1e-5 (clip.py:64-81); bias-free visual_projection (clip.py:82-90);
  * vision_heads = vision_width // 64 (clip.py:60);
  * text tower: causal transformer, quickgelu, eps 1e-5, EOT pooling matmul'd
    against the bias-free text projection (clip.py:92-123,148-167);
  * logits = exp(logit_scale) * norm(img) @ norm(txt).T (clip.py:180-188).
"""

from __future__ import annotations

import math

import torch
from torch import nn

from jimm_amd.models.common.text import TextTransformer
from jimm_amd.models.common.vit import VisionTransformerBase


class CLIP(nn.Module):
    def __init__(
        self,
        embed_dim: int = 512,
        image_resolution: int = 224,
        vision_layers: int = 12,
        vision_width: int = 768,
        vision_patch_size: int = 32,
        context_length: int = 77,
        vocab_size: int = 49408,
        transformer_width: int = 512,
        transformer_heads: int = 8,
        transformer_layers: int = 12,
        vision_mlp_dim: int | None = None,
        transformer_mlp_dim: int | None = None,
        *,
        hidden_act: str = "quickgelu",  # OpenAI CLIP; LAION variants use "gelu"
        layernorm_epsilon: float = 1e-5,
    ) -> None:
        super().__init__()
        self.embed_dim = embed_dim
        self.context_length = context_length
        vision_heads = max(1, vision_width // 64)  # clip.py:60
        self.vision_model = VisionTransformerBase(
            img_size=image_resolution,
            patch_size=vision_patch_size,
            hidden_size=vision_width,
            num_layers=vision_layers,
            num_heads=vision_heads,
            mlp_dim=vision_mlp_dim or vision_width * 4,
            use_pre_norm=True,
            use_patch_bias=False,
            hidden_act=hidden_act,
            pooling="CLS",
            layernorm_epsilon=layernorm_epsilon,
        )
        self.visual_projection = nn.Linear(vision_width, embed_dim, bias=False)
        self.text_model = TextTransformer(
            vocab_size=vocab_size,
            context_length=context_length,
            hidden_size=transformer_width,
            num_layers=transformer_layers,
            num_heads=transformer_heads,
            mlp_dim=transformer_mlp_dim or transformer_width * 4,
            causal=True,
            pooling="EOT",
            hidden_act=hidden_act,
            layernorm_epsilon=layernorm_epsilon,
        )
        self.text_projection = nn.Linear(transformer_width, embed_dim, bias=False)
        self.logit_scale = nn.Parameter(torch.tensor(math.log(1 / 0.07)))

    def gradient_checkpointing_enable(self, every_n: int = 1) -> None:
        """Recompute encoder blocks in backward; every_n > 1 checkpoints
        only every n-th block (selective)."""
        self.vision_model.encoder.gradient_checkpointing = True
        self.text_model.encoder.gradient_checkpointing = True
        self.vision_model.encoder.checkpoint_every = every_n
        self.text_model.encoder.checkpoint_every = every_n

    def encode_image(self, images: torch.Tensor) -> torch.Tensor:
        from jimm_amd.parallel.tp import row_parallel_linear

        return row_parallel_linear(self.vision_model(images), self.visual_projection)

    def encode_text(self, input_ids: torch.Tensor) -> torch.Tensor:
        from jimm_amd.parallel.tp import row_parallel_linear

        return row_parallel_linear(self.text_model(input_ids), self.text_projection)

    def forward(self, images: torch.Tensor, input_ids: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """Returns (logits_per_image, logits_per_text) — clip.py:169-188."""
        img = self.encode_image(images)
        txt = self.encode_text(input_ids)
        img = img / img.norm(dim=-1, keepdim=True)
        txt = txt / txt.norm(dim=-1, keepdim=True)
        scale = self.logit_scale.exp()
        logits_per_image = scale * img @ txt.t()
        return logits_per_image, logits_per_image.t()

    @classmethod
    def from_pretrained(cls, model_name_or_path: str, *, use_pytorch: bool = False, dtype: torch.dtype = torch.float32, device: str | torch.device = "cpu") -> "CLIP":
        from jimm_amd.interop.clip_hf import load_clip

        return load_clip(cls, model_name_or_path, use_pytorch=use_pytorch, dtype=dtype, device=device)

    def save_pretrained(self, save_dir: str) -> None:
        from jimm_amd.interop.clip_hf import save_clip

        save_clip(self, save_dir)
