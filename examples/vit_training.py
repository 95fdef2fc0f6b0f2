"""ViT classification training — the MI355X-native counterpart of the
reference's examples/vit_training.py (MNIST DP training loop,
/root/reference/examples/vit_training.py).

Trains a small ViT on sklearn's offline digits dataset (no network needed;
the reference uses tfds MNIST). Runs on CPU or GPU; with torchrun and N
processes it data-parallels over RCCL/xGMI:

    python examples/vit_training.py                       # single process
    torchrun --standalone --nproc-per-node 8 examples/vit_training.py
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch
import torch.nn.functional as F

import jimm_amd
from jimm_amd.train import Meter, TrainConfig, Trainer, init_distributed

BATCH = 64
LR = 1e-4          # reference: Adam 1e-4 (vit_training.py:202)
IMG = 32           # digits 8x8 upscaled


def load_digits_batches(batch, rank, world, seed=0):
    from sklearn.datasets import load_digits

    d = load_digits()
    imgs = torch.tensor(d.images, dtype=torch.float32) / 16.0  # (1797, 8, 8)
    imgs = F.interpolate(imgs.unsqueeze(1), size=(IMG, IMG), mode="bilinear")
    imgs = imgs.repeat(1, 3, 1, 1)  # grey -> 3 channels
    labels = torch.tensor(d.target)
    n = len(labels)
    g = torch.Generator().manual_seed(seed)
    perm = torch.randperm(n, generator=g)
    train_idx, test_idx = perm[: int(0.8 * n)], perm[int(0.8 * n) :]
    return (imgs[train_idx], labels[train_idx]), (imgs[test_idx], labels[test_idx])


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=5)
    ap.add_argument("--jsonl", default=None, help="metrics JSONL output path")
    args = ap.parse_args()
    rank, world, local_rank, device = init_distributed()
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    (tr_x, tr_y), (te_x, te_y) = load_digits_batches(BATCH, rank, world)

    torch.manual_seed(0)
    model = jimm_amd.VisionTransformer(
        num_classes=10, img_size=IMG, patch_size=8, num_layers=2,
        hidden_size=128, num_heads=2, mlp_dim=256,
    ).to(device, dtype)
    trainer = Trainer(model, TrainConfig(task="vit", lr=LR))
    meter = Meter(args.jsonl, rank=rank)

    n_train = len(tr_y)
    steps_per_epoch = n_train // (BATCH * world)
    g = torch.Generator().manual_seed(1)
    step = 0
    for epoch in range(args.epochs):
        perm = torch.randperm(n_train, generator=g)
        for i in range(steps_per_epoch):
            idx = perm[(i * world + rank) * BATCH : (i * world + rank + 1) * BATCH]
            batch = (tr_x[idx].to(device, dtype), tr_y[idx].to(device))
            out = trainer.train_step(batch)
            if step % 10 == 0:
                meter.log(step, loss=out["loss"].item(), acc=out["accuracy"].item(), epoch=epoch)
            step += 1
        # eval
        with torch.no_grad():
            logits = model(te_x.to(device, dtype))
            acc = (logits.argmax(-1).cpu() == te_y).float().mean().item()
        meter.log(step, test_acc=acc, epoch=epoch)
    if rank == 0:
        print(f"final test accuracy: {acc:.4f}")


if __name__ == "__main__":
    main()
