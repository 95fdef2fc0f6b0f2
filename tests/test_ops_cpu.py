"""CPU reference-path op tests (these ops are the oracle the HIP kernels
are tested against in tests/test_kernels_gpu.py)."""

import math

import torch
import torch.nn.functional as F

from jimm_amd import ops


def test_quickgelu():
    x = torch.randn(64)
    assert torch.allclose(ops.quickgelu(x), x * torch.sigmoid(1.702 * x))


def test_layer_norm_matches_torch():
    x = torch.randn(4, 7, 32)
    w, b = torch.randn(32), torch.randn(32)
    for eps in (1e-12, 1e-6, 1e-5):
        assert torch.allclose(ops.layer_norm(x, w, b, eps), F.layer_norm(x, (32,), w, b, eps), atol=1e-6)


def test_attention_matches_sdpa():
    q = torch.randn(2, 3, 11, 16)
    k = torch.randn(2, 3, 13, 16)
    v = torch.randn(2, 3, 13, 16)
    out = ops.attention(q, k, v)
    ref = F.scaled_dot_product_attention(q, k, v)
    assert torch.allclose(out, ref, atol=1e-5)


def test_attention_causal():
    q = k = v = torch.randn(1, 2, 9, 8)
    out = ops.attention(q, k, v, causal=True)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True)
    assert torch.allclose(out, ref, atol=1e-5)
    # causality: changing a future key must not change earlier outputs
    k2 = k.clone()
    k2[:, :, -1] += 10.0
    out2 = ops.attention(q, k2, v, causal=True)
    assert torch.allclose(out[:, :, :-1], out2[:, :, :-1], atol=1e-6)


def test_linear_epilogues():
    x = torch.randn(5, 12)
    w = torch.randn(8, 12)
    b = torch.randn(8)
    res = torch.randn(5, 8)
    assert torch.allclose(ops.linear(x, w, b), F.linear(x, w, b), atol=1e-6)
    assert torch.allclose(ops.linear(x, w, b, act="gelu"), F.gelu(F.linear(x, w, b)), atol=1e-6)
    assert torch.allclose(ops.linear(x, w, b, act="gelu_tanh"), F.gelu(F.linear(x, w, b), approximate="tanh"), atol=1e-6)
    z = F.linear(x, w, b)
    assert torch.allclose(ops.linear(x, w, b, act="quickgelu"), z * torch.sigmoid(1.702 * z), atol=1e-6)
    assert torch.allclose(ops.linear(x, w, b, residual=res), F.linear(x, w, b) + res, atol=1e-6)


def test_patch_embed_matches_conv():
    img = torch.randn(2, 3, 32, 32)
    w = torch.randn(16, 3, 8, 8)
    b = torch.randn(16)
    y = ops.patch_embed(img, w, b, 8)
    ref = F.conv2d(img, w, b, stride=8).flatten(2).transpose(1, 2)
    assert y.shape == (2, 16, 16)
    assert torch.allclose(y, ref, atol=1e-5)


def test_add_cls_pos():
    x = torch.randn(2, 4, 8)
    cls = torch.randn(1, 1, 8)
    pos = torch.randn(1, 5, 8)
    y = ops.add_cls_pos(x, cls, pos)
    assert y.shape == (2, 5, 8)
    assert torch.allclose(y[:, 0], cls[0, 0] + pos[0, 0])
    assert torch.allclose(y[:, 1:], x + pos[:, 1:])
    # MAP mode: no cls token
    y2 = ops.add_cls_pos(x, None, pos)
    assert torch.allclose(y2, x + pos[:, :4])


def test_attention_grad():
    q = torch.randn(1, 2, 5, 4, requires_grad=True)
    k = torch.randn(1, 2, 5, 4, requires_grad=True)
    v = torch.randn(1, 2, 5, 4, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda q, k, v: ops.attention(q.double(), k.double(), v.double()),
        (q.double(), k.double(), v.double()),
        eps=1e-6,
        atol=1e-4,
    )


def test_trainer_checkpoint_resume(tmp_path):
    """Trainer state (params + Adam moments + step) round-trips exactly."""
    import jimm_amd
    from jimm_amd.train import SyntheticImages, TrainConfig, Trainer

    torch.manual_seed(0)
    def make():
        torch.manual_seed(1)
        m = jimm_amd.VisionTransformer(num_classes=5, img_size=32, patch_size=16,
                                       num_layers=1, num_heads=2, mlp_dim=64, hidden_size=32)
        return Trainer(m, TrainConfig(task="vit", lr=1e-3))

    data = SyntheticImages(2, 32, 5, torch.device("cpu"), seed=3)
    it = iter(data)
    batches = [next(it) for _ in range(4)]

    tr = make()
    for b in batches[:2]:
        tr.train_step(b)
    ck = str(tmp_path / "ck.pt")
    tr.save_checkpoint(ck)
    for b in batches[2:]:
        tr.train_step(b)
    ref = [p.detach().clone() for p in tr.model.parameters()]

    tr2 = make()
    tr2.load_checkpoint(ck)
    assert tr2.step_idx == 2
    for b in batches[2:]:
        tr2.train_step(b)
    for p, q in zip(ref, tr2.model.parameters()):
        assert torch.allclose(p, q, atol=1e-6), (p - q).abs().max()


def test_eval_step_contrastive():
    import jimm_amd
    from jimm_amd.train import TrainConfig, Trainer

    torch.manual_seed(0)
    m = jimm_amd.CLIP(embed_dim=16, image_resolution=32, vision_layers=1, vision_width=64,
                      vision_patch_size=16, context_length=5, vocab_size=33,
                      transformer_width=32, transformer_heads=2, transformer_layers=1)
    tr = Trainer(m, TrainConfig(task="clip"))
    out = tr.eval_step((torch.randn(4, 3, 32, 32), torch.randint(0, 33, (4, 5))))
    assert set(out) == {"retrieval_i2t", "retrieval_t2i"}
    assert 0.0 <= out["retrieval_i2t"].item() <= 1.0


def test_gradient_checkpointing_matches():
    """Checkpointed blocks (full and selective every-n) give the same
    loss/grads as the plain path."""
    import jimm_amd

    torch.manual_seed(0)
    def run(ckpt, every_n=1):
        torch.manual_seed(1)
        m = jimm_amd.VisionTransformer(num_classes=5, img_size=32, patch_size=16,
                                       num_layers=4, num_heads=2, mlp_dim=64, hidden_size=32)
        if ckpt:
            m.gradient_checkpointing_enable(every_n=every_n)
        x = torch.randn(2, 3, 32, 32)
        out = m(x)
        out.square().sum().backward()
        return out.detach(), [p.grad.clone() for p in m.parameters()]

    o1, g1 = run(False)
    for every_n in (1, 2, 3):
        o2, g2 = run(True, every_n)
        assert torch.allclose(o1, o2, atol=1e-6), every_n
        for a, b in zip(g1, g2):
            assert torch.allclose(a, b, atol=1e-5), every_n


def test_lr_schedule():
    import jimm_amd
    from jimm_amd.train import SyntheticImages, TrainConfig, Trainer

    torch.manual_seed(0)
    m = jimm_amd.VisionTransformer(num_classes=5, img_size=32, patch_size=16,
                                   num_layers=1, num_heads=2, mlp_dim=64, hidden_size=32)
    tr = Trainer(m, TrainConfig(task="vit", lr=1e-3, lr_schedule="cosine",
                                warmup_steps=2, total_steps=10, min_lr=1e-5))
    data = SyntheticImages(2, 32, 5, torch.device("cpu"))
    it = iter(data)
    lrs = []
    for _ in range(10):
        tr.train_step(next(it))
        lrs.append(tr.opt.param_groups[0]["lr"])
    assert lrs[0] < lrs[1] <= 1e-3          # warmup ramps
    assert lrs[-1] < lrs[3]                 # cosine decays
    assert lrs[-1] >= 1e-5


def test_siglip_loss_chunking_invariant():
    """Column-chunked sigmoid loss equals the unchunked computation."""
    from jimm_amd.ops import losses as L

    torch.manual_seed(0)
    img = torch.randn(16, 32, requires_grad=True)
    txt = torch.randn(16, 32, requires_grad=True)
    s, b = torch.tensor(0.5), torch.tensor(-1.0)
    l1 = L.siglip_sigmoid_loss(img, txt, s, b, gather=False, chunk_size=5)
    l2 = L.siglip_sigmoid_loss(img, txt, s, b, gather=False, chunk_size=1 << 20)
    assert torch.allclose(l1, l2, atol=1e-5), (l1, l2)


def test_clip_loss_gather_flag_world1():
    """With no process group, gather=True must equal gather=False."""
    from jimm_amd.ops import losses as L

    torch.manual_seed(0)
    img = torch.randn(8, 16)
    txt = torch.randn(8, 16)
    s = torch.tensor(0.3)
    a = L.clip_contrastive_loss(img, txt, s, gather=True)
    b = L.clip_contrastive_loss(img, txt, s, gather=False)
    assert torch.allclose(a, b)


def test_synthetic_text_eot_invariant():
    """CLIP pools at argmax(ids): the synthetic pipeline must place exactly
    one EOS (= max id) per row so EOT pooling is well-defined."""
    from jimm_amd.train import SyntheticImageText

    data = SyntheticImageText(8, 32, 12, 100, torch.device("cpu"), seed=4)
    _, ids = next(iter(data))
    assert (ids == 99).sum(dim=1).eq(1).all()
    assert ids.max() == 99


def test_meter_jsonl(tmp_path):
    import json

    from jimm_amd.train import Meter

    p = tmp_path / "m.jsonl"
    m = Meter(str(p), rank=0)
    m.log(1, loss=0.5, acc=0.9)
    m.log(2, loss=0.4)
    m.close()
    recs = [json.loads(l) for l in open(p)]
    assert recs[0]["loss"] == 0.5 and recs[1]["step"] == 2
