"""Soak: 500 graph-replayed ViT-B train steps — loss must fall, memory flat."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import jimm_amd
from jimm_amd.ops._backend import maybe_enable_tunableop
from jimm_amd.train import SyntheticImages, TrainConfig, Trainer

maybe_enable_tunableop()
dev = torch.device("cuda:0")
torch.manual_seed(0)
model = jimm_amd.VisionTransformer(num_classes=1000).to(dev, torch.bfloat16)
tr = Trainer(model, TrainConfig(task="vit", lr=3e-4))
# fixed small label space makes the synthetic task learnable
data = SyntheticImages(256, 224, 8, dev, dtype=torch.bfloat16)
it = iter(data)
tr.enable_graph(next(it))
losses = []
t0 = time.perf_counter()
STEPS = int(os.environ.get("SOAK_STEPS", "500"))
for step in range(STEPS):
    out = tr.train_step(next(it))
    if step % 100 == 0 or step == STEPS - 1:
        torch.cuda.synchronize()
        losses.append(out["loss"].item())
        print(f"step {step}: loss {losses[-1]:.4f} mem {torch.cuda.memory_allocated()/2**30:.2f} GiB "
              f"peak {torch.cuda.max_memory_allocated()/2**30:.2f} GiB", flush=True)
dt = time.perf_counter() - t0
print(f"{STEPS} steps in {dt:.1f}s ({STEPS*256/dt:.0f} img/s); loss {losses[0]:.3f} -> {losses[-1]:.3f}")
assert losses[-1] < losses[0], "loss did not decrease"
print("SOAK OK")
