"""Training loop — the first-class trainer the reference keeps in an example
(/root/reference/examples/vit_training.py:60-102,178-236).

``train_step`` semantics preserved: forward -> loss (+accuracy for
classification) -> backward -> Adam update; gradients DP-all-reduced over
RCCL, overlapped with backward (parallel/ddp.py). Logging sync is periodic
(the reference's per-step ``.item()`` forces a device sync every step —
vit_training.py:222-223 — which we deliberately avoid).
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.distributed as dist

from jimm_amd.ops import losses as L
from jimm_amd.parallel.ddp import DataParallelGrads
from jimm_amd.train.adam import Adam


@dataclass
class TrainConfig:
    task: str = "vit"  # "vit" | "clip" | "siglip"
    lr: float = 1e-4
    betas: tuple = (0.9, 0.999)
    weight_decay: float = 0.0
    bucket_bytes: int = 25 * 1024 * 1024
    log_every: int = 50
    # LR schedule: None = constant; "cosine" decays lr -> min_lr over
    # total_steps after a linear warmup_steps ramp
    lr_schedule: str | None = None
    warmup_steps: int = 0
    total_steps: int = 0
    min_lr: float = 0.0
    extra: dict = field(default_factory=dict)


class Trainer:
    def __init__(self, model: torch.nn.Module, cfg: TrainConfig, *, process_group=None):
        from jimm_amd.ops._backend import maybe_enable_tunableop

        maybe_enable_tunableop()  # committed hipBLASLt algo table (MI355X)
        self.model = model
        self.cfg = cfg
        self.ddp = DataParallelGrads(model, bucket_bytes=cfg.bucket_bytes, process_group=process_group)
        self.opt = Adam(model.parameters(), lr=cfg.lr, betas=cfg.betas, weight_decay=cfg.weight_decay)
        self.group = process_group
        self.step_idx = 0
        # JIMM_AMD_ROCTX=1: wrap each step in a roctx range (shows up in
        # rocprofv3 --marker-trace; SURVEY §5 tracing)
        import os

        self._roctx = os.environ.get("JIMM_AMD_ROCTX", "0") == "1"
        self._graph = None
        self._static_batch = None
        self._static_out = None

    def _current_lr(self) -> float:
        c = self.cfg
        if c.lr_schedule is None:
            return c.lr
        step = self.step_idx
        if c.warmup_steps and step < c.warmup_steps:
            return c.lr * (step + 1) / c.warmup_steps
        if c.lr_schedule == "cosine" and c.total_steps:
            import math

            t = min(max(step - c.warmup_steps, 0), max(c.total_steps - c.warmup_steps, 1))
            f = 0.5 * (1 + math.cos(math.pi * t / max(c.total_steps - c.warmup_steps, 1)))
            return c.min_lr + (c.lr - c.min_lr) * f
        return c.lr

    def _apply_lr(self) -> None:
        lr = self._current_lr()
        for g in self.opt.param_groups:
            g["lr"] = lr
        # under graph replay the captured Adam reads lr from device memory:
        # refresh it OUTSIDE the graph
        if self._graph is not None and getattr(self.opt, "_prepared", None):
            self.opt.set_device_lr(lr)

    def train_step(self, batch) -> dict[str, torch.Tensor]:
        if self.cfg.lr_schedule is not None:
            self._apply_lr()
        if self._roctx:
            torch.cuda.nvtx.range_push(f"train_step_{self.cfg.task}")  # roctx range on ROCm
        if self._graph is not None:
            out = self._graph_step(batch)
        else:
            out = self._train_step_inner(batch)
        if self._roctx:
            torch.cuda.nvtx.range_pop()
        return out

    # -- hipGraph capture of the whole step (fwd+bwd+allreduce+Adam) --------
    # ~1300 kernel launches/step left a ~5 ms launch gap in the round-1
    # profiles; capture once, then one graph replay per step. Synthetic/real
    # batches are copied into static input buffers OUTSIDE the graph so every
    # step still computes on fresh data.
    def enable_graph(self, example_batch, warmup: int = 3) -> None:
        assert torch.cuda.is_available(), "graph capture needs a GPU"
        self._static_batch = tuple(t.clone() for t in example_batch)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                self._static_out = self._train_step_inner(self._static_batch)
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._static_out = self._train_step_inner(self._static_batch)
        self._graph = g

    def _graph_step(self, batch):
        for dst, src in zip(self._static_batch, batch):
            dst.copy_(src, non_blocking=True)
        self._graph.replay()
        self.step_idx += 1
        return self._static_out

    def _train_step_inner(self, batch) -> dict[str, torch.Tensor]:
        self.ddp.zero_grad()
        out: dict[str, torch.Tensor] = {}
        if self.cfg.task == "vit":
            images, labels = batch
            logits = self.model(images)
            loss = L.softmax_cross_entropy(logits, labels)
            out["accuracy"] = (logits.argmax(-1) == labels).float().mean()
        elif self.cfg.task == "clip":
            images, ids = batch
            img_emb = self.model.encode_image(images)
            txt_emb = self.model.encode_text(ids)
            loss = L.clip_contrastive_loss(img_emb, txt_emb, self.model.logit_scale, group=self.group)
        elif self.cfg.task == "siglip":
            images, ids = batch
            img_emb = self.model.encode_image(images)
            txt_emb = self.model.encode_text(ids)
            loss = L.siglip_sigmoid_loss(img_emb, txt_emb, self.model.logit_scale, self.model.logit_bias, group=self.group)
        else:
            raise ValueError(self.cfg.task)
        loss.backward()
        self.ddp.finalize()
        self.opt.step()
        self.step_idx += 1
        out["loss"] = loss.detach()
        return out

    def comm_stats(self) -> dict:
        """Per-step collective traffic (per-collective bytes).

        allreduce_bytes counts the DP gradient payload actually reduced in
        the latest step; gather_bytes the contrastive-loss embedding
        all-gather (clip/siglip tasks)."""
        stats = {
            "allreduce_bytes": self.ddp.last_step_bytes,
            "allreduce_buckets": len(self.ddp.buckets),
            "grad_bytes_total": self.ddp.grad_bytes(),
        }
        from jimm_amd.parallel import gather as _g

        stats["gather_bytes"] = getattr(_g, "_last_gather_bytes", 0)
        return stats

    # -- checkpoint / resume (the reference is load-only; we add
    # full trainer state so long runs are resumable) ------------------------
    def save_checkpoint(self, path: str) -> None:
        """Rank-0 writes model + Adam state (moments, masters, step count)."""
        import torch.distributed as dist

        if dist.is_initialized() and dist.get_rank(self.group) != 0:
            return
        if hasattr(self.opt, "sync_step_from_device"):
            self.opt.sync_step_from_device()
        torch.save(
            {
                "model": self.model.state_dict(),
                "optimizer": self.opt.state_dict(),
                "step_idx": self.step_idx,
                "task": self.cfg.task,
            },
            path,
        )

    def load_checkpoint(self, path: str) -> None:
        ckpt = torch.load(path, map_location="cpu", weights_only=False)
        self.model.load_state_dict(ckpt["model"])
        self.opt.load_state_dict(ckpt["optimizer"])
        # moments/masters must live on the parameters' device
        dev = next(self.model.parameters()).device
        for st in self.opt.state.values():
            for k, v in st.items():
                if torch.is_tensor(v):
                    st[k] = v.to(dev)
        self.step_idx = ckpt["step_idx"]

    @torch.no_grad()
    def eval_step(self, batch) -> dict[str, torch.Tensor]:
        if self.cfg.task == "vit":
            images, labels = batch
            logits = self.model(images)
            return {
                "loss": L.softmax_cross_entropy(logits, labels),
                "accuracy": (logits.argmax(-1) == labels).float().mean(),
            }
        # clip/siglip: local-batch image->text retrieval accuracy (diagonal)
        images, ids = batch
        logits_per_image, _ = self.model(images, ids)
        labels = torch.arange(logits_per_image.shape[0], device=logits_per_image.device)
        return {
            "retrieval_i2t": (logits_per_image.argmax(-1) == labels).float().mean(),
            "retrieval_t2i": (logits_per_image.argmax(0) == labels).float().mean(),
        }


def init_distributed(device_type: str | None = None):
    """Initialize torch.distributed from torchrun env vars; no-op single-proc.

    Returns (rank, world_size, local_rank, device).
    """
    import os

    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
        if device.type == "cuda":
            torch.cuda.set_device(device)
        return 0, 1, 0, device
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
        backend = "nccl"  # RCCL on ROCm
    else:
        device = torch.device("cpu")
        backend = "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    return rank, world, local_rank, device
