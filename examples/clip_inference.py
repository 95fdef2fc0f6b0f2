"""CLIP zero-shot classification — counterpart of the reference's
examples/clip_inference.py (6 prompts vs 1 image, softmax ranking;
/root/reference/examples/clip_inference.py:17-52).

Pass a local HF checkpoint dir (e.g. openai/clip-vit-base-patch32 cloned
offline) or run with random-init weights as a smoke demo."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import jimm_amd

PROMPTS = [
    "a photo of a cat", "a photo of a dog", "a photo of a car",
    "a photo of a tree", "a photo of a house", "a photo of a bird",
]


def simple_tokenize(texts, context_length=77, vocab=49408):
    """Hash-based stand-in tokenizer for the offline demo (real use: pass
    checkpoint dir + `transformers` CLIPTokenizer)."""
    ids = torch.zeros(len(texts), context_length, dtype=torch.long)
    for i, t in enumerate(texts):
        toks = [hash(w) % (vocab - 2) for w in t.split()]
        ids[i, : len(toks)] = torch.tensor(toks)
        ids[i, len(toks)] = vocab - 1  # EOT = max id (CLIP pools at argmax)
    return ids


def main():
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    if len(sys.argv) > 1:
        model = jimm_amd.CLIP.from_pretrained(sys.argv[1], dtype=torch.float32).to(device, dtype)
        try:
            from transformers import CLIPTokenizer

            tok = CLIPTokenizer.from_pretrained(sys.argv[1])
            ids = tok(PROMPTS, padding="max_length", max_length=77, return_tensors="pt").input_ids
        except Exception:
            ids = simple_tokenize(PROMPTS)
    else:
        torch.manual_seed(0)
        model = jimm_amd.CLIP().to(device, dtype)
        ids = simple_tokenize(PROMPTS)
    model.eval()

    img = torch.randn(1, 3, 224, 224, device=device, dtype=dtype)
    with torch.no_grad():
        logits_per_image, _ = model(img, ids.to(device))
        probs = logits_per_image.float().softmax(-1)[0]
    for p, prob in sorted(zip(PROMPTS, probs.tolist()), key=lambda x: -x[1]):
        print(f"{prob:6.3f}  {p}")


if __name__ == "__main__":
    main()
