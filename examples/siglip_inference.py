"""SigLIP inference smoke — counterpart of the reference's
examples/siglip_inference.ipynb (random-init SigLIP + encode_image run).

Pass a local HF checkpoint dir for real weights; defaults to random-init
google/siglip-base-patch16-256 dims."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd


def main():
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    torch.manual_seed(0)
    if len(sys.argv) > 1:
        model = jimm_amd.SigLIP.from_pretrained(sys.argv[1], dtype=torch.float32).to(device, dtype)
    else:
        model = jimm_amd.SigLIP()  # base-patch16-256 dims, random init
        model = model.to(device, dtype)
    model.eval()

    imgs = torch.randn(2, 3, 256, 256, device=device, dtype=dtype)
    ids = torch.randint(0, 32000, (2, 64), device=device)
    with torch.no_grad():
        img_emb = model.encode_image(imgs)        # MAP-head output
        txt_emb = model.encode_text(ids)
        logits_per_image, _ = model(imgs, ids)
    print("image embedding:", tuple(img_emb.shape), "text embedding:", tuple(txt_emb.shape))
    print("logits_per_image:\n", logits_per_image.float().cpu())


if __name__ == "__main__":
    main()
