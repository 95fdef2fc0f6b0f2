"""Flagship benchmark — BASELINE.json metric: images/sec (whole node) for
ViT-B/16@224 bf16 training (default), plus CLIP-B/32 pairs/sec (--task clip)
and SigLIP-base/16-256 (--task siglip).

Driver contract: `python bench.py --gpus N --steps K --warmup W`; for N>1 the
driver launches via torch.distributed.run with one rank per GPU over RCCL.
W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize() on both sides; elapsed is the MAX over
ranks; rank 0 prints ONE JSON line.

Synthetic data (no network), random-init weights, bf16 compute on GPU.
"""

from __future__ import annotations

import argparse
import json
import time

import torch
import torch.distributed as dist


def build(task: str, device, dtype):
    import jimm_amd
    from jimm_amd.train import SyntheticImages, SyntheticImageText, TrainConfig, Trainer

    if task == "vit":
        model = jimm_amd.VisionTransformer(num_classes=1000, img_size=224, patch_size=16)  # ViT-B/16
        model_name = "ViT-B/16@224"
    elif task == "vitl384":
        # BASELINE config 5: ViT-L/16 @ 384 with multihead-attention pooling
        model = jimm_amd.VisionTransformer(
            num_classes=1000, img_size=384, patch_size=16, num_layers=24,
            hidden_size=1024, num_heads=16, mlp_dim=4096, pooling="MAP",
        )
        model_name = "ViT-L/16@384-MAP"
    elif task == "clip":
        model = jimm_amd.CLIP(  # CLIP ViT-B/32 (openai/clip-vit-base-patch32 dims)
            embed_dim=512,
            image_resolution=224,
            vision_layers=12,
            vision_width=768,
            vision_patch_size=32,
            context_length=77,
            vocab_size=49408,
            transformer_width=512,
            transformer_heads=8,
            transformer_layers=12,
        )
        model_name = "CLIP-B/32"
    elif task == "siglip":
        model = jimm_amd.SigLIP(  # google/siglip-base-patch16-256 dims
            image_resolution=256,
            vision_layers=12,
            vision_width=768,
            vision_patch_size=16,
            context_length=64,
            vocab_size=32000,
            transformer_width=768,
            transformer_heads=12,
            transformer_layers=12,
        )
        model_name = "SigLIP-base/16-256"
    else:
        raise ValueError(task)
    model = model.to(device=device, dtype=dtype)
    # LayerNorm/scalars stay bf16-safe; master fp32 weights live in Adam state
    trainer = Trainer(model, TrainConfig(task="vit" if task.startswith("vit") else task))
    return model, trainer, model_name


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--task", choices=["vit", "vitl384", "clip", "siglip"], default="vit")
    p.add_argument("--dtype", choices=["bf16", "fp8"], default="bf16",
                   help="fp8: e4m3 forward GEMMs via MFMA fp8 (bf16 backward)")
    p.add_argument("--batch", type=int, default=None, help="per-GPU batch size")
    p.add_argument("--ckpt", action="store_true", help="gradient checkpointing (huge batches)")
    p.add_argument("--graph", choices=["auto", "1", "0"], default="auto",
                   help="hipGraph-capture the train step (any world size; capture "
                        "failure falls back to eager COORDINATED across ranks)")
    args = p.parse_args()

    from jimm_amd.ops._backend import maybe_enable_tunableop
    from jimm_amd.train import SyntheticImages, SyntheticImageText
    from jimm_amd.train.trainer import init_distributed

    maybe_enable_tunableop()  # JIMM_AMD_TUNABLE=<csv>: committed hipBLASLt algo table

    rank, world, local_rank, device = init_distributed()
    on_gpu = device.type == "cuda"
    dtype = torch.bfloat16 if on_gpu else torch.float32
    batch = args.batch or ({"vit": 1024, "vitl384": 128, "clip": 1024, "siglip": 512}[args.task] if on_gpu else 4)

    torch.manual_seed(1234 + rank)
    model, trainer, model_name = build(args.task, device, dtype)
    if args.ckpt:
        model.gradient_checkpointing_enable()

    if args.dtype == "fp8" and on_gpu:
        from jimm_amd.ops import set_fp8

        set_fp8(True)
    if args.task in ("vit", "vitl384"):
        img = 224 if args.task == "vit" else 384
        data = SyntheticImages(batch, img, 1000, device, dtype=dtype, seed=rank)
        items_per_step = batch * world  # images
        unit = "images/sec"
    else:
        img_size = 224 if args.task == "clip" else 256
        ctx = 77 if args.task == "clip" else 64
        vocab = 49408 if args.task == "clip" else 32000
        data = SyntheticImageText(batch, img_size, ctx, vocab, device, dtype=dtype, seed=rank)
        items_per_step = batch * world  # image-text pairs
        unit = "pairs/sec"

    def sync():
        if dist.is_initialized():
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    it = iter(data)
    # hipGraph-capture the whole train step (RCCL collectives capture too —
    # verified by benchmarks/graph_nccl_probe.py). Fallback is COORDINATED:
    # if any rank fails to capture, every rank runs eager.
    use_graph = on_gpu and args.graph != "0"
    if use_graph:
        ok = True
        try:
            trainer.enable_graph(next(it))
        except Exception as e:
            ok = False
            import sys

            if rank == 0:
                print(f"# graph capture failed ({e!r}); running eager", file=sys.stderr, flush=True)
        if dist.is_initialized():
            flag = torch.tensor([1 if ok else 0], dtype=torch.int32, device=device)
            dist.all_reduce(flag, op=dist.ReduceOp.MIN)
            if flag.item() == 0:
                trainer._graph = None  # someone failed: everyone eager
    for _ in range(args.warmup):
        trainer.train_step(next(it))
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        trainer.train_step(next(it))
    sync()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if on_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    if rank == 0:
        value = items_per_step * args.steps / elapsed
        out = {
            "metric": f"{unit} (whole node) {model_name} train",
            "value": value,
            "unit": unit,
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no throughput numbers
            "dtype": (args.dtype if on_gpu else "fp32"),
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": items_per_step,
                "img_size": {"vit": 224, "vitl384": 384, "clip": 224, "siglip": 256}[args.task],
                "parallelism": f"dp{world}",
            },
        }
        print(json.dumps(out), flush=True)
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
