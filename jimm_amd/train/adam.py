"""Adam optimizer with a fused HIP multi-tensor kernel on GPU (K14).

Matches optax.adam semantics used by the reference trainer
(/root/reference/examples/vit_training.py:202-203): bias-corrected first and
second moments, no weight decay by default (AdamW-style decoupled decay
available via ``weight_decay``).

Mixed precision: parameters may be bf16; moments (and optional fp32 master
weights) are kept in fp32. On GPU the whole update runs in one fused HIP
kernel launch per bucket of tensors (csrc/adam.hip); the CPU path is the
numerics oracle.
"""

from __future__ import annotations

import torch

from jimm_amd.ops import _backend


class Adam(torch.optim.Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-4,
        betas: tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        *,
        master_weights: bool = True,
    ) -> None:
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.master_weights = master_weights
        # graph-capturable fast path: chunk descriptors uploaded once, lr and
        # step live on-device (csrc/adam.hip adam_prepare/adam_apply).
        # Keyed per param group: each group gets its own descriptor table,
        # lr and device-side step counter.
        self._prepared: dict[int, tuple] = {}
        self._last_lr: dict[int, float] = {}

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            lr, (b1, b2), eps, wd = group["lr"], group["betas"], group["eps"], group["weight_decay"]
            ps, gs, ms, vs, masters = [], [], [], [], []
            step_t = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if not st:
                    st["step"] = 0
                    st["m"] = torch.zeros_like(p, dtype=torch.float32)
                    st["v"] = torch.zeros_like(p, dtype=torch.float32)
                    if self.master_weights and p.dtype != torch.float32:
                        st["master"] = p.detach().clone().float()
                st["step"] += 1
                step_t = st["step"]
                ps.append(p)
                gs.append(p.grad)
                ms.append(st["m"])
                vs.append(st["v"])
                masters.append(st.get("master"))
            if not ps:
                continue
            if ps[0].is_cuda and not _backend.force_eager():
                ext = _backend.ext()
                if gi not in self._prepared:
                    if step_t == 1:
                        # first step initializes state lazily; take it eagerly
                        # then freeze the descriptor table (pointers stable)
                        ext.adam_step(ps, gs, ms, vs, masters, lr, b1, b2, eps, wd, step_t)
                    desc, nchunks = ext.adam_prepare(ps, gs, ms, vs, masters)
                    lr_dev = torch.full((1,), lr, dtype=torch.float32, device=ps[0].device)
                    # adam_apply pre-increments on device: seed so the next
                    # apply computes bias correction for the right step
                    seed = step_t if step_t == 1 else step_t - 1
                    step_dev = torch.full((1,), seed, dtype=torch.int32, device=ps[0].device)
                    self._prepared[gi] = (desc, int(nchunks.item()), lr_dev, step_dev)
                    if step_t == 1:
                        continue
                desc, nchunks, lr_dev, step_dev = self._prepared[gi]
                if lr != self._last_lr.get(gi):
                    lr_dev.fill_(lr)
                    self._last_lr[gi] = lr
                ext.adam_apply(desc, nchunks, lr_dev, step_dev, b1, b2, eps, wd)
            else:
                self._ref_step(ps, gs, ms, vs, masters, lr, b1, b2, eps, wd, step_t)
        return loss

    def set_device_lr(self, lr: float) -> None:
        """Refresh the device-side lr used by graph-replayed adam_apply."""
        for gi, prep in self._prepared.items():
            prep[2].fill_(lr)
            self._last_lr[gi] = lr

    def load_state_dict(self, state_dict) -> None:
        super().load_state_dict(state_dict)
        # The base class casts loaded state to the param dtype; for bf16
        # params that would silently demote the fp32 moments/master weights
        # (and break the fused kernel's data_ptr<float>()). Restore fp32.
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state.get(p)
                if not st:
                    continue
                for key in ("m", "v", "master"):
                    if key in st and st[key] is not None and st[key].dtype != torch.float32:
                        st[key] = st[key].float()
        self._prepared = {}  # moment tensors were replaced: descriptors stale
        self._last_lr = {}

    @torch.no_grad()
    def sync_step_from_device(self) -> None:
        """Copy the on-device step counter back into per-param state (needed
        after graph-replayed training, where host counters do not advance)."""
        if not self._prepared:
            return
        for gi, group in enumerate(self.param_groups):
            if gi not in self._prepared:
                continue
            step = int(self._prepared[gi][3].item())
            for p in group["params"]:
                if self.state[p]:
                    self.state[p]["step"] = step

    @staticmethod
    def _ref_step(ps, gs, ms, vs, masters, lr, b1, b2, eps, wd, t):
        bc1 = 1.0 - b1**t
        bc2 = 1.0 - b2**t
        for p, g, m, v, master in zip(ps, gs, ms, vs, masters):
            gf = g.float()
            m.mul_(b1).add_(gf, alpha=1 - b1)
            v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
            mhat = m / bc1
            vhat = v / bc2
            upd = mhat / (vhat.sqrt() + eps)
            target = master if master is not None else p
            if wd != 0.0:
                upd = upd + wd * target.float()
            target.add_(upd.to(target.dtype), alpha=-lr)
            if master is not None:
                p.copy_(master.to(p.dtype))
