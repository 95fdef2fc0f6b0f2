"""ViT image classifier.

Reference: /root/reference/src/jimm/models/vit.py:16-273.
Tower configuration preserved: CLS pooling, no pre-norm, patch bias,
exact-gelu, layernorm eps 1e-12 (vit.py:61-78); optional classification head
gated by ``do_classification`` (vit.py:80-89,100-103). Defaults are
ViT-B/16 @ 224 (vit.py:25-33).
"""

from __future__ import annotations

import torch
from torch import nn

from jimm_amd import ops
from jimm_amd.models.common.vit import VisionTransformerBase


class VisionTransformer(nn.Module):
    def __init__(
        self,
        num_classes: int = 1000,
        in_channels: int = 3,
        img_size: int = 224,
        patch_size: int = 16,
        num_layers: int = 12,
        num_heads: int = 12,
        mlp_dim: int = 3072,
        hidden_size: int = 768,
        dropout_rate: float = 0.0,
        *,
        do_classification: bool = True,
        hidden_act: str = "gelu",
        layernorm_epsilon: float = 1e-12,
        pooling: str = "CLS",  # "CLS" | "MAP" (BASELINE config 5: ViT-L/384 MAP)
    ) -> None:
        super().__init__()
        self.do_classification = do_classification
        self.vision = VisionTransformerBase(
            img_size=img_size,
            patch_size=patch_size,
            in_channels=in_channels,
            hidden_size=hidden_size,
            num_layers=num_layers,
            num_heads=num_heads,
            mlp_dim=mlp_dim,
            dropout_rate=dropout_rate,
            use_pre_norm=False,
            use_patch_bias=True,
            hidden_act=hidden_act,
            pooling=pooling,
            layernorm_epsilon=layernorm_epsilon,
        )
        if do_classification:
            self.classifier = nn.Linear(hidden_size, num_classes, bias=True)
        else:
            self.classifier = None

    def gradient_checkpointing_enable(self, every_n: int = 1) -> None:
        """Recompute encoder blocks in backward; every_n > 1 checkpoints
        only every n-th block (selective: 1/n the recompute cost for a
        partial memory win)."""
        self.vision.encoder.gradient_checkpointing = True
        self.vision.encoder.checkpoint_every = every_n

    def forward(self, images: torch.Tensor) -> torch.Tensor:
        x = self.vision(images)  # (B, H) CLS- or MAP-pooled
        if self.do_classification:
            x = ops.linear(x, self.classifier.weight, self.classifier.bias)
        return x

    @classmethod
    def from_pretrained(cls, model_name_or_path: str, *, use_pytorch: bool = False, dtype: torch.dtype = torch.float32, device: str | torch.device = "cpu") -> "VisionTransformer":
        from jimm_amd.interop.vit_hf import load_vit

        return load_vit(cls, model_name_or_path, use_pytorch=use_pytorch, dtype=dtype, device=device)

    def save_pretrained(self, save_dir: str) -> None:
        from jimm_amd.interop.vit_hf import save_vit

        save_vit(self, save_dir)
