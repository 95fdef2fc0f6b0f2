"""Two identical 5-step runs (same seed) must produce bitwise-identical
parameters under JIMM_AMD_DETERMINISTIC=1."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import jimm_amd
from jimm_amd.train import SyntheticImages, TrainConfig, Trainer

def run():
    torch.manual_seed(0)
    m = jimm_amd.VisionTransformer(num_classes=100).to("cuda", torch.bfloat16)
    tr = Trainer(m, TrainConfig(task="vit", lr=1e-3))
    data = SyntheticImages(64, 224, 100, torch.device("cuda"), dtype=torch.bfloat16, seed=9)
    it = iter(data)
    for _ in range(5):
        tr.train_step(next(it))
    torch.cuda.synchronize()
    return [p.detach().clone() for p in m.parameters()]

a = run()
b = run()
bad = [i for i, (p, q) in enumerate(zip(a, b)) if not torch.equal(p, q)]
print("mismatched params:", len(bad), "of", len(a))
assert not bad, bad[:5]
print("DETERMINISM OK")
