"""Multi-process logic tests on CPU (gloo, world_size 2) — SURVEY §4
implication (c)/(d): bucketed DP all-reduce, all-gather-with-grad, and the
DP-invariance property (loss/grads at DP=2 equal single-process on the same
global batch)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import jimm_amd
from jimm_amd.ops import losses as L

pytestmark = pytest.mark.dist


def _run(rank, world, fn, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    try:
        fn(rank, world)
    finally:
        dist.destroy_process_group()


def spawn(fn, world=2, port=29511):
    mp.spawn(_run, args=(world, fn, port), nprocs=world, join=True)


# ---------------------------------------------------------------------------
def _check_allgather_grad(rank, world):
    from jimm_amd.parallel.gather import all_gather_with_grad

    x = torch.randn(3, 4, requires_grad=True)  # same seed -> same on each rank
    local = x * (rank + 1.0)
    g = all_gather_with_grad(local)
    assert g.shape == (3 * world, 4)
    # loss uses EVERY rank's slice differently
    w = torch.arange(1.0, 1.0 + g.shape[0]).unsqueeze(1)
    (g * w).sum().backward()
    # d/d local_r = sum over ranks of w-slice-r (each rank computes the same
    # loss here) -> grad = world * w_r ; then * (rank+1) chain to x
    wr = w.view(world, 3, 1)[rank]
    expect = world * wr.expand(3, 4) * (rank + 1.0)
    assert torch.allclose(x.grad, expect, atol=1e-5), (rank, x.grad, expect)


def test_allgather_with_grad():
    spawn(_check_allgather_grad, port=29512)


# ---------------------------------------------------------------------------
def _check_ddp_average(rank, world):
    from jimm_amd.parallel.ddp import DataParallelGrads

    m = torch.nn.Linear(4, 3)
    ddp = DataParallelGrads(m, bucket_bytes=8)  # force multiple buckets
    data = lambda r: torch.arange(20.0).reshape(5, 4) * (r + 1.0) / 10.0
    ddp.zero_grad()
    m(data(rank)).sum().backward()
    ddp.finalize()
    got = [p.grad.clone() for p in m.parameters()]
    # reference: average of per-rank grads
    m2 = torch.nn.Linear(4, 3)
    m2.load_state_dict(m.state_dict())
    ref = None
    for r in range(world):
        m2.zero_grad()
        m2(data(r)).sum().backward()
        g = [p.grad.clone() for p in m2.parameters()]
        ref = g if ref is None else [a + b for a, b in zip(ref, g)]
    ref = [g / world for g in ref]
    for a, b in zip(got, ref):
        assert torch.allclose(a, b, atol=1e-5)


def test_ddp_bucketed_average():
    spawn(_check_ddp_average, port=29513)


# ---------------------------------------------------------------------------
def _dp_invariance(rank, world):
    """CLIP loss + grads at DP=2 must equal the single-process global batch."""
    torch.manual_seed(0)
    model = jimm_amd.CLIP(embed_dim=16, image_resolution=32, vision_layers=1, vision_width=64,
                          vision_patch_size=16, context_length=8, vocab_size=100,
                          transformer_width=64, transformer_heads=1, transformer_layers=1)
    from jimm_amd.parallel.ddp import DataParallelGrads

    ddp = DataParallelGrads(model)  # also broadcasts params
    torch.manual_seed(42)
    imgs = torch.randn(4, 3, 32, 32)
    ids = torch.randint(0, 99, (4, 8))
    ids[:, -1] = 99
    # shard the global batch
    b = 4 // world
    my_imgs, my_ids = imgs[rank * b : (rank + 1) * b], ids[rank * b : (rank + 1) * b]
    ddp.zero_grad()
    img_e = model.encode_image(my_imgs)
    txt_e = model.encode_text(my_ids)
    loss = L.clip_contrastive_loss(img_e, txt_e, model.logit_scale)
    loss.backward()
    ddp.finalize()

    # single-process reference on the full batch
    model_ref = jimm_amd.CLIP(embed_dim=16, image_resolution=32, vision_layers=1, vision_width=64,
                              vision_patch_size=16, context_length=8, vocab_size=100,
                              transformer_width=64, transformer_heads=1, transformer_layers=1)
    model_ref.load_state_dict(model.state_dict())
    img_r = model_ref.encode_image(imgs)
    txt_r = model_ref.encode_text(ids)
    loss_r = L.clip_contrastive_loss(img_r, txt_r, model_ref.logit_scale, gather=False)
    loss_r.backward()

    # per-rank loss differs (local rows) but the DP-averaged loss matches
    lt = loss.detach().clone()
    dist.all_reduce(lt)
    assert torch.allclose(lt / world, loss_r.detach(), atol=1e-5), (lt / world, loss_r)
    for (n, p), (nr, pr) in zip(model.named_parameters(), model_ref.named_parameters()):
        assert n == nr
        if pr.grad is None:
            continue
        assert torch.allclose(p.grad, pr.grad, atol=1e-4), f"{n}: {(p.grad - pr.grad).abs().max()}"


def test_dp_invariance_clip():
    spawn(_dp_invariance, port=29514)


# ---------------------------------------------------------------------------
def _siglip_invariance(rank, world):
    torch.manual_seed(0)
    img = torch.randn(6, 16)
    txt = torch.randn(6, 16)
    scale = torch.tensor(0.7, requires_grad=True)
    bias = torch.tensor(-1.0, requires_grad=True)
    b = 6 // world
    my_img = img[rank * b : (rank + 1) * b].clone().requires_grad_(True)
    my_txt = txt[rank * b : (rank + 1) * b].clone().requires_grad_(True)
    loss = L.siglip_sigmoid_loss(my_img, my_txt, scale, bias, chunk_size=4)
    loss.backward()
    # reference single-process
    img_r = img.clone().requires_grad_(True)
    txt_r = txt.clone().requires_grad_(True)
    scale_r = torch.tensor(0.7, requires_grad=True)
    bias_r = torch.tensor(-1.0, requires_grad=True)
    loss_r = L.siglip_sigmoid_loss(img_r, txt_r, scale_r, bias_r, gather=False)
    loss_r.backward()
    lt = loss.detach().clone()
    dist.all_reduce(lt)
    assert torch.allclose(lt / world, loss_r.detach(), atol=1e-5)
    # DP grad average must equal the reference grads for this rank's shard
    gi = my_img.grad.clone()
    gt = my_txt.grad.clone()
    dist.all_reduce(gi)  # emb grads are local (no param sharing) -> no averaging;
    # instead compare directly: ref grad on shard r vs (local grad / world)?
    # The DP convention averages PARAM grads; embedding inputs here stand in
    # for activations: expected relation is my_img.grad == world * ref-shard
    # after loss-mean-normalization (loss normalized by B_local, ref by B_global)
    assert torch.allclose(my_img.grad / world, img_r.grad[rank * b : (rank + 1) * b], atol=1e-5)
    assert torch.allclose(my_txt.grad / world, txt_r.grad[rank * b : (rank + 1) * b], atol=1e-5)
    # scale/bias: DP-average of per-rank grads == reference grad
    st = scale.grad.clone()
    dist.all_reduce(st)
    assert torch.allclose(st / world, scale_r.grad, atol=1e-5)


def test_dp_invariance_siglip():
    spawn(_siglip_invariance, port=29515)


# ---------------------------------------------------------------------------
def _check_tensor_parallel(rank, world):
    from jimm_amd.models.common.transformer import EncoderBlock
    from jimm_amd.parallel.tp import shard_encoder_block

    torch.manual_seed(0)
    blk = EncoderBlock(64, 4, 128, hidden_act="gelu", layernorm_epsilon=1e-6)
    x = torch.randn(2, 9, 64)
    dy = torch.randn(2, 9, 64)

    ref_blk = EncoderBlock(64, 4, 128, hidden_act="gelu", layernorm_epsilon=1e-6)
    ref_blk.load_state_dict(blk.state_dict())
    xr = x.clone().requires_grad_(True)
    yr = ref_blk(xr)
    yr.backward(dy)

    shard_encoder_block(blk, None)
    xt = x.clone().requires_grad_(True)
    yt = blk(xt)
    yt.backward(dy)

    assert torch.allclose(yt, yr, atol=1e-5), (yt - yr).abs().max()
    assert torch.allclose(xt.grad, xr.grad, atol=1e-5), (xt.grad - xr.grad).abs().max()
    mrows = slice(rank * (128 // world), (rank + 1) * (128 // world))
    assert torch.allclose(blk.fc1.weight.grad, ref_blk.fc1.weight.grad[mrows], atol=1e-5)
    assert torch.allclose(blk.norm1.weight.grad, ref_blk.norm1.weight.grad, atol=1e-5)


def test_tensor_parallel_block():
    spawn(_check_tensor_parallel, port=29515)


# ---------------------------------------------------------------------------
def _tiny_clip():
    # vision_width 256 -> 4 vision heads (width // 64): shardable at tp=4
    return jimm_amd.CLIP(embed_dim=16, image_resolution=32, vision_layers=1, vision_width=256,
                         vision_patch_size=16, context_length=8, vocab_size=103,
                         transformer_width=64, transformer_heads=4, transformer_layers=1)


def _check_full_model_tp_clip(rank, world):
    """the WHOLE CLIP model tensor-sharded (encoders +
    vocab-parallel embedding + row-parallel projections) matches the
    unsharded oracle exactly — forward logits and input grads."""
    from jimm_amd.parallel.tp import shard_clip

    torch.manual_seed(0)
    model = _tiny_clip()
    ref = _tiny_clip()
    ref.load_state_dict(model.state_dict())

    g = torch.Generator().manual_seed(3)
    imgs = torch.randn(2, 3, 32, 32, generator=g)
    ids = torch.randint(0, 102, (2, 8), generator=g)
    ids[:, -1] = 102  # EOT

    imgs_r = imgs.clone().requires_grad_(True)
    lr, _ = ref(imgs_r, ids)
    lr.sum().backward()

    shard_clip(model, None)
    imgs_t = imgs.clone().requires_grad_(True)
    lt, _ = model(imgs_t, ids)
    lt.sum().backward()

    assert torch.allclose(lt, lr, atol=1e-4), (lt - lr).abs().max()
    assert torch.allclose(imgs_t.grad, imgs_r.grad, atol=1e-4)
    # vocab-parallel embedding grads match the oracle's local rows
    per = (103 + world - 1) // world
    s, e = rank * per, min((rank + 1) * per, 103)
    assert torch.allclose(
        model.text_model.token_embedding.weight.grad,
        ref.text_model.token_embedding.weight.grad[s:e], atol=1e-4,
    )
    # row-parallel projection grads match the oracle's column slice
    per_c = 256 // world
    assert torch.allclose(
        model.visual_projection.weight.grad,
        ref.visual_projection.weight.grad[:, rank * per_c:(rank + 1) * per_c], atol=1e-4,
    )


def test_full_model_tp_clip_world4():
    spawn(_check_full_model_tp_clip, world=4, port=29517)


def _check_full_model_tp_siglip(rank, world):
    from jimm_amd.parallel.tp import shard_siglip

    def make():
        return jimm_amd.SigLIP(image_resolution=32, vision_layers=1, vision_width=64,
                               vision_patch_size=16, context_length=8, vocab_size=50,
                               transformer_width=64, transformer_heads=4,
                               transformer_layers=1, vision_heads=4)  # noqa

    torch.manual_seed(0)
    model = make()
    ref = make()
    ref.load_state_dict(model.state_dict())

    g = torch.Generator().manual_seed(5)
    imgs = torch.randn(2, 3, 32, 32, generator=g)
    ids = torch.randint(0, 50, (2, 8), generator=g)

    imgs_r = imgs.clone().requires_grad_(True)
    lr, _ = ref(imgs_r, ids)
    lr.sum().backward()

    shard_siglip(model, None)
    imgs_t = imgs.clone().requires_grad_(True)
    lt, _ = model(imgs_t, ids)
    lt.sum().backward()

    assert torch.allclose(lt, lr, atol=1e-4), (lt - lr).abs().max()
    assert torch.allclose(imgs_t.grad, imgs_r.grad, atol=1e-4)


def test_full_model_tp_siglip_world4():
    spawn(_check_full_model_tp_siglip, world=4, port=29518)


def _check_full_model_tp_vit(rank, world):
    from jimm_amd.parallel.tp import shard_vit

    def make():
        return jimm_amd.VisionTransformer(num_classes=7, img_size=32, patch_size=16,
                                          num_layers=2, num_heads=4, mlp_dim=128,
                                          hidden_size=64, pooling="MAP")

    torch.manual_seed(0)
    model = make()
    ref = make()
    ref.load_state_dict(model.state_dict())

    g = torch.Generator().manual_seed(7)
    imgs = torch.randn(2, 3, 32, 32, generator=g)

    imgs_r = imgs.clone().requires_grad_(True)
    lr = ref(imgs_r)
    lr.sum().backward()

    shard_vit(model, None)
    imgs_t = imgs.clone().requires_grad_(True)
    lt = model(imgs_t)
    lt.sum().backward()

    assert torch.allclose(lt, lr, atol=1e-4), (lt - lr).abs().max()
    assert torch.allclose(imgs_t.grad, imgs_r.grad, atol=1e-4)


def test_full_model_tp_vit_world4():
    """ViT (MAP pooling) full-model TP vs the unsharded oracle — completes
    the TP test matrix (CLIP/SigLIP covered above)."""
    spawn(_check_full_model_tp_vit, world=4, port=29528)


# ---------------------------------------------------------------------------
def _check_trainer_dp_invariance(rank, world):
    """Full Trainer step at DP=2 matches a single-process run on the same
    global batch (covers bucketed all-reduce + Adam together)."""
    import jimm_amd
    from jimm_amd.train import TrainConfig, Trainer

    torch.manual_seed(0)
    def make():
        torch.manual_seed(42)
        m = jimm_amd.VisionTransformer(num_classes=5, img_size=32, patch_size=16,
                                       num_layers=1, num_heads=2, mlp_dim=64, hidden_size=32)
        return m

    g = torch.Generator().manual_seed(7)
    imgs = torch.randn(4, 3, 32, 32, generator=g)
    labels = torch.randint(0, 5, (4,), generator=g)

    # single-process oracle on the global batch
    ref = make()
    ref_tr = Trainer.__new__(Trainer)  # build without DDP broadcast
    from jimm_amd.parallel.ddp import DataParallelGrads
    from jimm_amd.train.adam import Adam

    ref_tr.model = ref
    ref_tr.cfg = TrainConfig(task="vit", lr=1e-3)
    ref_tr.ddp = DataParallelGrads.__new__(DataParallelGrads)
    ref_tr.ddp.enabled = False
    ref_tr.ddp.buckets = []
    ref_tr.ddp.zero_grad = lambda: ref.zero_grad(set_to_none=False)
    ref_tr.ddp.finalize = lambda: None
    ref_tr.opt = Adam(ref.parameters(), lr=1e-3)
    ref_tr.group = None
    ref_tr.step_idx = 0
    ref_tr._roctx = False
    ref_tr._graph = None
    for _ in range(3):
        ref_tr.train_step((imgs, labels))

    # DP run: each rank takes its half of the global batch
    model = make()
    tr = Trainer(model, TrainConfig(task="vit", lr=1e-3))
    lo, hi = rank * 2, rank * 2 + 2
    for _ in range(3):
        tr.train_step((imgs[lo:hi], labels[lo:hi]))

    # Adam's first-step update is sign(g)*lr, so reduction-order noise on
    # near-zero grads flips whole +-lr quanta — compare at the lr scale
    for p, q in zip(ref.parameters(), model.parameters()):
        assert torch.allclose(p, q, atol=5e-3), (p - q).abs().max()


def test_trainer_dp_invariance():
    spawn(_check_trainer_dp_invariance, port=29517)


def test_ddp_average_world4():
    spawn(_check_ddp_average, world=4, port=29519)


def test_tensor_parallel_world4():
    spawn(_check_tensor_parallel, world=4, port=29521)


# ---------------------------------------------------------------------------
def _trainer_world8(rank, world):
    """Trainer DP-invariance at gloo world 8 (harden for the
    first real 8-GPU run) — loss average over 8 ranks equals the
    single-process loss on the same global batch after 2 steps."""
    from jimm_amd.train import TrainConfig, Trainer

    def make():
        torch.manual_seed(42)
        return jimm_amd.VisionTransformer(num_classes=5, img_size=32, patch_size=16,
                                          num_layers=1, num_heads=2, mlp_dim=64, hidden_size=32)

    g = torch.Generator().manual_seed(7)
    imgs = torch.randn(8, 3, 32, 32, generator=g)
    labels = torch.randint(0, 5, (8,), generator=g)

    ref = make()
    from jimm_amd.train.adam import Adam

    opt = Adam(ref.parameters(), lr=1e-3)
    ref_losses = []
    for _ in range(2):
        ref.zero_grad(set_to_none=False)
        import jimm_amd.ops.losses as LL

        loss = LL.softmax_cross_entropy(ref(imgs), labels)
        loss.backward()
        opt.step()
        ref_losses.append(loss.detach())

    tr = Trainer(make(), TrainConfig(task="vit", lr=1e-3))
    my = (imgs[rank:rank + 1], labels[rank:rank + 1])
    for i in range(2):
        out = tr.train_step(my)
        lt = out["loss"].detach().clone()
        dist.all_reduce(lt)
        assert torch.allclose(lt / world, ref_losses[i], atol=1e-4), (i, lt / world, ref_losses[i])


def test_trainer_dp_invariance_world8():
    spawn(_trainer_world8, world=8, port=29519)


# ---------------------------------------------------------------------------
def _bucket_overlap_stress(rank, world):
    """Bucket-overlap stress: tiny buckets force MANY
    concurrent async all-reduces with interleaved completion; grads must
    still match the serial average, repeatedly (catches reuse-before-wait
    and pending-counter bugs)."""
    from jimm_amd.parallel.ddp import DataParallelGrads

    torch.manual_seed(0)
    m = torch.nn.Sequential(*[torch.nn.Linear(16, 16) for _ in range(12)])
    ddp = DataParallelGrads(m, bucket_bytes=64)  # ~dozens of buckets
    assert len(ddp.buckets) >= 12, len(ddp.buckets)
    for it in range(5):
        x = torch.randn(4, 16, generator=torch.Generator().manual_seed(100 + it))
        ddp.zero_grad()
        m(x * (rank + 1.0)).pow(2).sum().backward()
        ddp.finalize()
        # serial reference
        m2 = torch.nn.Sequential(*[torch.nn.Linear(16, 16) for _ in range(12)])
        m2.load_state_dict(m.state_dict())
        ref = None
        for r in range(world):
            m2.zero_grad()
            m2(x * (r + 1.0)).pow(2).sum().backward()
            gs = [p.grad.clone() for p in m2.parameters()]
            ref = gs if ref is None else [a + b for a, b in zip(ref, gs)]
        for p, rg in zip(m.parameters(), ref):
            assert torch.allclose(p.grad, rg / world, atol=1e-4)


def _comm_stats_world2(rank, world):
    """Trainer.comm_stats surfaces the per-step collective traffic
    (SURVEY §5 metrics): all-reduce bytes equal the full grad payload and
    the contrastive gather bytes are nonzero for clip-task steps."""
    import jimm_amd
    from jimm_amd.train import SyntheticImageText, TrainConfig, Trainer

    torch.manual_seed(0)
    m = jimm_amd.CLIP(
        embed_dim=16, image_resolution=32, vision_layers=1, vision_width=32,
        vision_patch_size=16, context_length=8, vocab_size=64,
        transformer_width=32, transformer_heads=2, transformer_layers=1,
    ).float()
    tr = Trainer(m, TrainConfig(task="clip", lr=1e-3))
    data = SyntheticImageText(2, 32, 8, 64, torch.device("cpu"), dtype=torch.float32, seed=rank)
    tr.train_step(next(iter(data)))
    st = tr.comm_stats()
    assert st["allreduce_bytes"] == st["grad_bytes_total"] > 0, st
    assert st["allreduce_buckets"] >= 1
    assert st["gather_bytes"] > 0, st  # image+text embedding all-gathers


def test_comm_stats_world2():
    spawn(_comm_stats_world2, port=29527)


def test_bench_torchrun_world2():
    """The driver's exact launch path: torchrun --nproc-per-node 2 bench.py
    on CPU/gloo — rendezvous on 127.0.0.1, per-rank data shards, barrier +
    max-over-ranks timing, exactly one JSON line from rank 0."""
    import json
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "2"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.join(os.path.dirname(__file__), ".."),
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, lines  # exactly one JSON line (rank 0)
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2 and out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 4  # 2 ranks x per-rank batch 2
    assert out["value"] > 0 and out["steps"] == 2
    # full driver contract: every key the round harness parses
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    for key in ("model", "global_batch", "img_size", "parallelism"):
        assert key in out["config"], key
    assert out["scaling"] == "weak" and out["higher_is_better"] is True
    assert out["data"] == "synthetic"


def test_bucket_overlap_stress():
    spawn(_bucket_overlap_stress, world=4, port=29520)
