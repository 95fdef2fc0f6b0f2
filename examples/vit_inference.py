"""ViT-L/16-384 bf16 batched inference throughput — counterpart of the
reference's examples/vit_inference.py (128 batches x 128 images, bf16,
jit-once-and-reuse; /root/reference/examples/vit_inference.py:14-63)."""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd

N_BATCHES = 16
BATCH = 128


def main():
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    torch.manual_seed(0)
    model = jimm_amd.VisionTransformer(  # ViT-L/16 @ 384
        num_classes=1000, img_size=384, patch_size=16,
        num_layers=24, hidden_size=1024, num_heads=16, mlp_dim=4096,
    ).to(device, dtype).eval()

    imgs = torch.randn(BATCH, 3, 384, 384, device=device, dtype=dtype)
    with torch.no_grad():
        model(imgs)  # warmup
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(N_BATCHES):
            model(imgs)
        if device.type == "cuda":
            torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{N_BATCHES * BATCH / dt:.1f} images/sec  ({dt / N_BATCHES * 1e3:.1f} ms/batch of {BATCH})")


if __name__ == "__main__":
    main()
