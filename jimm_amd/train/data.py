"""Synthetic data pipelines (there is no network for datasets — BASELINE.json:
synthetic 224²/384² image/caption batches, random-init weights).

Batches are generated directly on the device, double-buffered so generation
overlaps the training step (matters only for huge batches; generation is a
single RNG kernel).
"""

from __future__ import annotations

import torch


class SyntheticImages:
    """Endless (images, labels) batches of fixed shape."""

    def __init__(self, batch_size: int, img_size: int, num_classes: int, device, dtype=torch.float32, seed: int = 0):
        self.batch_size, self.img_size, self.num_classes = batch_size, img_size, num_classes
        self.device, self.dtype = device, dtype
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(seed)

    def __iter__(self):
        return self

    def __next__(self):
        imgs = torch.randn(self.batch_size, 3, self.img_size, self.img_size, generator=self.gen, device=self.device, dtype=self.dtype)
        labels = torch.randint(0, self.num_classes, (self.batch_size,), generator=self.gen, device=self.device)
        return imgs, labels


class SyntheticImageText:
    """Endless (images, input_ids) batches for CLIP/SigLIP training."""

    def __init__(self, batch_size: int, img_size: int, context_length: int, vocab_size: int, device, dtype=torch.float32, seed: int = 0, eos_id: int | None = None):
        self.batch_size, self.img_size = batch_size, img_size
        self.context_length, self.vocab_size = context_length, vocab_size
        self.device, self.dtype = device, dtype
        self.eos_id = eos_id if eos_id is not None else vocab_size - 1
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(seed)

    def __iter__(self):
        return self

    def __next__(self):
        imgs = torch.randn(self.batch_size, 3, self.img_size, self.img_size, generator=self.gen, device=self.device, dtype=self.dtype)
        # ids < eos everywhere, one EOS per row (CLIP pools at argmax == EOS)
        ids = torch.randint(0, self.eos_id, (self.batch_size, self.context_length), generator=self.gen, device=self.device)
        pos = torch.randint(1, self.context_length, (self.batch_size,), generator=self.gen, device=self.device)
        ids.scatter_(1, pos.unsqueeze(1), self.eos_id)
        return imgs, ids
