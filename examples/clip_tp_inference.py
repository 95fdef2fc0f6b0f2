"""Tensor-parallel CLIP inference — the counterpart of the reference's
examples/clip_inference.py, which runs with mesh (1, n_devices) =
("batch","model"), i.e. PURE tensor sharding of the weights across all
devices (/root/reference/examples/clip_inference.py:17-18).

Here the sharding is explicit (parallel/tp.py shard_clip) and covers the
WHOLE model: every encoder block keeps num_heads/N heads and mlp_dim/N of
the MLP per rank, the token embedding is vocab-parallel, and the
visual/text projections are row-parallel — with all-reduces after each
row-parallel product.

    torchrun --standalone --nproc-per-node 8 examples/clip_tp_inference.py
    (also runs on CPU with gloo for any world size that divides the heads)
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd
from jimm_amd.parallel.tp import shard_clip
from jimm_amd.train.trainer import init_distributed

PROMPTS = [
    "a photo of a cat", "a photo of a dog", "a photo of a car",
    "a photo of a tree", "a photo of a house", "a photo of a bird",
]


def simple_tokenize(texts, context_length=77, vocab=49408):
    ids = torch.zeros(len(texts), context_length, dtype=torch.long)
    for i, t in enumerate(texts):
        toks = [hash(w) % (vocab - 2) for w in t.split()]
        ids[i, : len(toks)] = torch.tensor(toks)
        ids[i, len(toks)] = vocab - 1  # EOT
    return ids


def main():
    rank, world, local_rank, device = init_distributed()
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    torch.manual_seed(0)  # same seed on every rank -> replicated init
    if len(sys.argv) > 1:
        model = jimm_amd.CLIP.from_pretrained(sys.argv[1], dtype=torch.float32)
    else:
        model = jimm_amd.CLIP()
    model = model.to(device, dtype).eval()
    if world > 1:
        # FULL-model sharding: encoders + vocab-parallel token embedding +
        # row-parallel projections (the reference's (1, n) mesh shards every
        # parameter — clip.py:89,112-131)
        shard_clip(model, None)

    img = torch.randn(1, 3, 224, 224, device=device, dtype=dtype)
    ids = simple_tokenize(PROMPTS).to(device)
    with torch.no_grad():
        logits_per_image, _ = model(img, ids)
        probs = logits_per_image.float().softmax(-1)[0]
    if rank == 0:
        for p, prob in sorted(zip(PROMPTS, probs.tolist()), key=lambda x: -x[1]):
            print(f"{prob:6.3f}  {p}")


if __name__ == "__main__":
    main()
