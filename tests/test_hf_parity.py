"""Numerical-parity tests against HuggingFace transformers (in-process, no
network): tiny random-init HF models are saved with save_pretrained and
loaded through jimm_amd's from_pretrained, then outputs are compared.

This reproduces the reference's test strategy (SURVEY.md §4 —
tests/test_vit.py, test_clip.py, test_siglip.py) with offline fixtures and
much tighter tolerances (fp32, same-process): reference tolerances were
ViT max-abs < 0.05, CLIP atol 1e-1, SigLIP atol 1e-2.
"""

import numpy as np
import pytest
import torch

import jimm_amd

transformers = pytest.importorskip("transformers")


def _tiny_vit(tmp_path, old_keys: bool):
    from transformers import ViTConfig, ViTForImageClassification

    cfg = ViTConfig(
        hidden_size=32,
        num_hidden_layers=2,
        num_attention_heads=2,
        intermediate_size=64,
        image_size=32,
        patch_size=16,
        num_labels=3,
    )
    hf = ViTForImageClassification(cfg).eval()
    for p in hf.parameters():
        p.data.normal_(0, 0.02)
    d = tmp_path / "vit"
    hf.save_pretrained(d, safe_serialization=True)
    if old_keys:
        # rewrite to the classic hub key scheme (transformers <=4.x)
        from safetensors.torch import load_file, save_file

        sd = load_file(d / "model.safetensors")
        out = {}
        for k, t in sd.items():
            k = k.replace("vit.layers.", "vit.encoder.layer.")
            k = k.replace(".attention.q_proj.", ".attention.attention.query.")
            k = k.replace(".attention.k_proj.", ".attention.attention.key.")
            k = k.replace(".attention.v_proj.", ".attention.attention.value.")
            k = k.replace(".attention.o_proj.", ".attention.output.dense.")
            k = k.replace(".mlp.fc1.", ".intermediate.dense.")
            k = k.replace(".mlp.fc2.", ".output.dense.")
            out[k] = t
        save_file(out, str(d / "model.safetensors"))
    return hf, d


@pytest.mark.parametrize("old_keys", [False, True])
def test_vit_parity(tmp_path, old_keys):
    hf, d = _tiny_vit(tmp_path, old_keys)
    model = jimm_amd.VisionTransformer.from_pretrained(str(d)).eval()
    x = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        ref = hf(x).logits
        out = model(x)
    assert out.shape == ref.shape
    diff = (out - ref).abs().max().item()
    assert diff < 1e-4, f"max abs diff {diff}"  # reference bar: <0.05 (test_vit.py:49-52)


def _tiny_clip(tmp_path):
    from transformers import CLIPConfig, CLIPModel

    cfg = CLIPConfig(
        text_config=dict(
            hidden_size=32,
            intermediate_size=64,
            num_hidden_layers=2,
            num_attention_heads=2,
            max_position_embeddings=12,
            vocab_size=100,
            bos_token_id=98,
            eos_token_id=99,
        ),
        vision_config=dict(hidden_size=64, intermediate_size=128, num_hidden_layers=2, num_attention_heads=2, image_size=32, patch_size=16),
        projection_dim=16,
    )
    hf = CLIPModel(cfg).eval()
    for p in hf.parameters():
        p.data.normal_(0, 0.02)
    with torch.no_grad():
        hf.logit_scale.fill_(float(np.log(1 / 0.07)))
    d = tmp_path / "clip"
    hf.save_pretrained(d, safe_serialization=True)
    return hf, d


def test_clip_parity(tmp_path):
    hf, d = _tiny_clip(tmp_path)
    model = jimm_amd.CLIP.from_pretrained(str(d)).eval()
    img = torch.randn(2, 3, 32, 32)
    # ids: eos (=99, also the max id) at a known position per row
    ids = torch.randint(0, 98, (2, 12))
    ids[0, 5] = 99
    ids[1, 9] = 99
    with torch.no_grad():
        ref = hf(input_ids=ids, pixel_values=img)
        li, lt = model(img, ids)
    d_img = (li - ref.logits_per_image).abs().max().item()
    d_txt = (lt - ref.logits_per_text).abs().max().item()
    assert d_img < 1e-3 and d_txt < 1e-3, (d_img, d_txt)  # reference bar: atol 1e-1

    with torch.no_grad():
        out_i = model.encode_image(img)
        out_t = model.encode_text(ids)
    out_i = out_i / out_i.norm(dim=-1, keepdim=True)
    out_t = out_t / out_t.norm(dim=-1, keepdim=True)
    ref_i = ref.image_embeds  # HF returns the L2-normalized embeddings
    ref_t = ref.text_embeds
    assert (out_i - ref_i).abs().max().item() < 1e-5
    assert (out_t - ref_t).abs().max().item() < 1e-5


def _tiny_siglip(tmp_path):
    from transformers import SiglipConfig, SiglipModel

    cfg = SiglipConfig(
        text_config=dict(hidden_size=64, intermediate_size=96, num_hidden_layers=2, num_attention_heads=2, max_position_embeddings=12, vocab_size=100),
        vision_config=dict(hidden_size=64, intermediate_size=96, num_hidden_layers=2, num_attention_heads=2, image_size=32, patch_size=16),
    )
    hf = SiglipModel(cfg).eval()
    for p in hf.parameters():
        p.data.normal_(0, 0.02)
    d = tmp_path / "siglip"
    hf.save_pretrained(d, safe_serialization=True)
    return hf, d


def test_siglip_parity(tmp_path):
    hf, d = _tiny_siglip(tmp_path)
    model = jimm_amd.SigLIP.from_pretrained(str(d)).eval()
    img = torch.randn(2, 3, 32, 32)
    ids = torch.randint(0, 100, (2, 12))
    with torch.no_grad():
        ref = hf(input_ids=ids, pixel_values=img)
        out_i = model.encode_image(img)
        out_t = model.encode_text(ids)
        li, lt = model(img, ids)
    out_i = out_i / out_i.norm(dim=-1, keepdim=True)
    out_t = out_t / out_t.norm(dim=-1, keepdim=True)
    assert (out_i - ref.image_embeds).abs().max().item() < 1e-5  # reference bar: atol 1e-2
    assert (out_t - ref.text_embeds).abs().max().item() < 1e-5
    assert (li - ref.logits_per_image).abs().max().item() < 1e-3
    assert (lt - ref.logits_per_text).abs().max().item() < 1e-3


def test_vit_shape_inference(tmp_path):
    """Bare safetensors file (no config.json) -> shape-inferred config
    (reference models/vit.py:144-164)."""
    hf, d = _tiny_vit(tmp_path, old_keys=True)
    import os

    os.remove(d / "config.json")
    # heads inference assumes head_dim 64 (hidden//64) which fails for tiny
    # hidden sizes; check the structural fields instead by loading weights file
    weights = str(d / "model.safetensors")
    from jimm_amd.interop.vit_hf import _infer_config
    from safetensors.torch import load_file

    cfg = _infer_config(load_file(weights))
    assert cfg["hidden_size"] == 32
    assert cfg["num_layers"] == 2
    assert cfg["patch_size"] == 16
    assert cfg["img_size"] == 32
    assert cfg["mlp_dim"] == 64
    assert cfg["num_classes"] == 3


def test_siglip2_vision_parity(tmp_path):
    """SigLIP2 fixed-resolution checkpoints load through the same SigLIP
    class (reference supports 'SigLIP v1/v2', README.md:6-15): the linear
    channels-last patch embedding is reshaped to the conv layout."""
    from transformers import Siglip2Config, Siglip2Model, Siglip2TextConfig, Siglip2VisionConfig

    vc = Siglip2VisionConfig(hidden_size=32, num_hidden_layers=2, num_attention_heads=2,
                             intermediate_size=64, image_size=32, patch_size=16, num_patches=4)
    tc = Siglip2TextConfig(hidden_size=32, num_hidden_layers=2, num_attention_heads=2,
                           intermediate_size=64, vocab_size=99, max_position_embeddings=8,
                           bos_token_id=None, eos_token_id=None)
    hf = Siglip2Model(Siglip2Config(vision_config=vc.to_dict(), text_config=tc.to_dict())).eval()
    for p in hf.parameters():
        p.data.normal_(0, 0.02)
    d = tmp_path / "siglip2"
    hf.save_pretrained(d, safe_serialization=True)

    model = jimm_amd.SigLIP.from_pretrained(str(d)).eval()
    torch.manual_seed(0)
    img = torch.randn(2, 3, 32, 32)

    # oracle: pack the image into channels-last patches the Siglip2 way
    P = 16
    patches = img.permute(0, 2, 3, 1)                       # (B, H, W, C)
    patches = patches.reshape(2, 2, P, 2, P, 3).permute(0, 1, 3, 2, 4, 5).reshape(2, 4, P * P * 3)
    spatial = torch.tensor([[2, 2], [2, 2]])
    mask = torch.ones(2, 4, dtype=torch.long)
    with torch.no_grad():
        ref = hf.vision_model(pixel_values=patches, pixel_attention_mask=mask,
                              spatial_shapes=spatial).pooler_output
        out = model.encode_image(img)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()

    ids = torch.randint(0, 99, (2, 8))
    with torch.no_grad():
        ref_t = hf.text_model(input_ids=ids).pooler_output
        out_t = model.encode_text(ids)
    assert torch.allclose(out_t, ref_t, atol=1e-4), (out_t - ref_t).abs().max()


def test_checkpoint_coverage_asserts(tmp_path):
    """The bidirectional coverage asserts of the reference's from_pretrained
    (zero nonvisited, zero leftover, shape checks) fire."""
    from safetensors.torch import load_file, save_file

    hf, d = _tiny_vit(tmp_path, old_keys=False)
    sd_path = d / "model.safetensors"
    good = load_file(sd_path)

    # extra unconsumed key -> error
    bad = dict(good)
    bad["vit.mystery_weight"] = torch.zeros(3)
    save_file(bad, str(sd_path))
    with pytest.raises(ValueError, match="unconsumed"):
        jimm_amd.VisionTransformer.from_pretrained(str(d))

    # missing key -> error
    bad = {k: v for k, v in good.items() if "classifier.bias" not in k}
    save_file(bad, str(sd_path))
    with pytest.raises((ValueError, KeyError)):
        jimm_amd.VisionTransformer.from_pretrained(str(d))

    # position_ids buffers are tolerated (IGNORE_PATTERNS)
    ok = dict(good)
    ok["vit.embeddings.position_ids"] = torch.arange(5)
    save_file(ok, str(sd_path))
    jimm_amd.VisionTransformer.from_pretrained(str(d))


def test_roundtrip_save_load(tmp_path):
    """save_pretrained -> from_pretrained is the identity (all three models)."""
    torch.manual_seed(0)
    models = {
        "vit": jimm_amd.VisionTransformer(num_classes=3, img_size=32, patch_size=16,
                                          num_layers=1, num_heads=2, mlp_dim=64, hidden_size=32),
        "clip": jimm_amd.CLIP(embed_dim=16, image_resolution=32, vision_layers=1,
                              vision_width=64, vision_patch_size=16, context_length=5,
                              vocab_size=33, transformer_width=32, transformer_heads=2,
                              transformer_layers=1),
        "siglip": jimm_amd.SigLIP(image_resolution=32, vision_layers=1, vision_width=64,
                                  vision_patch_size=16, context_length=5, vocab_size=33,
                                  transformer_width=64, transformer_heads=1,
                                  transformer_layers=1, vision_heads=2),
    }
    for name, m in models.items():
        d = tmp_path / name
        m.save_pretrained(str(d))
        m2 = type(m).from_pretrained(str(d))
        for (k, a), (k2, b) in zip(sorted(m.state_dict().items()), sorted(m2.state_dict().items())):
            assert k == k2
            assert torch.equal(a, b), (name, k)


def test_roundtrip_fuzz_configs(tmp_path):
    """Randomized model configs survive save_pretrained -> from_pretrained
    exactly (hypothesis-style sweep without the decorator overhead)."""
    import random

    rng = random.Random(0)
    for trial in range(6):
        hidden = rng.choice([32, 64, 96])
        heads = rng.choice([1, 2]) if hidden % 64 else 2
        if hidden % heads:
            heads = 1
        layers = rng.choice([1, 2, 3])
        patch = rng.choice([8, 16])
        img = patch * rng.choice([2, 3, 4])
        ncls = rng.choice([2, 7, 10])
        m = jimm_amd.VisionTransformer(num_classes=ncls, img_size=img, patch_size=patch,
                                       num_layers=layers, num_heads=heads,
                                       mlp_dim=hidden * 2, hidden_size=hidden)
        d = tmp_path / f"t{trial}"
        m.save_pretrained(str(d))
        m2 = jimm_amd.VisionTransformer.from_pretrained(str(d))
        sd, sd2 = m.state_dict(), m2.state_dict()
        assert set(sd) == set(sd2)
        for k in sd:
            assert torch.equal(sd[k], sd2[k]), (trial, k)
        # and the forward agrees
        x = torch.randn(2, 3, img, img)
        with torch.no_grad():
            assert torch.allclose(m(x), m2(x), atol=1e-6)


def test_bare_safetensors_shape_inference(tmp_path):
    """A bare .safetensors file (no config.json) loads via shape inference
    (reference models/vit.py:144-164, clip.py:208-247, siglip.py:193-207)."""
    from safetensors.torch import load_file, save_file

    # ViT
    hf, d = _tiny_vit(tmp_path, old_keys=False)
    bare = tmp_path / "vit_bare.safetensors"
    save_file(load_file(d / "model.safetensors"), str(bare))
    m = jimm_amd.VisionTransformer.from_pretrained(str(bare)).eval()
    x = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        ref = hf(x).logits
        out = m(x)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()

    # CLIP and SigLIP roundtrip through a bare file
    clip = jimm_amd.CLIP(embed_dim=16, image_resolution=32, vision_layers=1,
                         vision_width=64, vision_patch_size=16, context_length=5,
                         vocab_size=33, transformer_width=64, transformer_heads=1,
                         transformer_layers=1)
    d2 = tmp_path / "clip"
    clip.save_pretrained(str(d2))
    bare2 = tmp_path / "clip_bare.safetensors"
    save_file(load_file(d2 / "model.safetensors"), str(bare2))
    clip2 = jimm_amd.CLIP.from_pretrained(str(bare2))
    for k, v in clip.state_dict().items():
        assert torch.equal(v, clip2.state_dict()[k]), k

    sig = jimm_amd.SigLIP(image_resolution=32, vision_layers=1, vision_width=64,
                          vision_patch_size=16, context_length=5, vocab_size=33,
                          transformer_width=64, transformer_heads=1,
                          transformer_layers=1, vision_heads=1)
    d3 = tmp_path / "sig"
    sig.save_pretrained(str(d3))
    bare3 = tmp_path / "sig_bare.safetensors"
    save_file(load_file(d3 / "model.safetensors"), str(bare3))
    sig2 = jimm_amd.SigLIP.from_pretrained(str(bare3))
    for k, v in sig.state_dict().items():
        assert torch.equal(v, sig2.state_dict()[k]), k


def test_pytorch_bin_format(tmp_path):
    """pytorch_model.bin checkpoints load via use_pytorch=True (the
    reference parametrizes safetensors vs torch-format — utils.py:56,70)."""
    from safetensors.torch import load_file

    hf, d = _tiny_vit(tmp_path, old_keys=False)
    sd = load_file(d / "model.safetensors")
    torch.save(sd, d / "pytorch_model.bin")
    (d / "model.safetensors").unlink()
    m = jimm_amd.VisionTransformer.from_pretrained(str(d), use_pytorch=True).eval()
    x = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        ref = hf(x).logits
        out = m(x)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


def test_clip_gelu_variant(tmp_path):
    """LAION-style CLIP configs (hidden_act='gelu') load with plain GELU
    instead of the OpenAI QuickGELU default."""
    from transformers import CLIPConfig, CLIPModel, CLIPTextConfig, CLIPVisionConfig

    vc = CLIPVisionConfig(hidden_size=64, num_hidden_layers=1, num_attention_heads=1,
                          intermediate_size=128, image_size=32, patch_size=16,
                          hidden_act="gelu")
    tc = CLIPTextConfig(hidden_size=32, num_hidden_layers=1, num_attention_heads=2,
                        intermediate_size=64, max_position_embeddings=6, vocab_size=50,
                        hidden_act="gelu", bos_token_id=0, eos_token_id=49)
    hf = CLIPModel(CLIPConfig(text_config=tc.to_dict(), vision_config=vc.to_dict(),
                              projection_dim=16)).eval()
    for p in hf.parameters():
        p.data.normal_(0, 0.02)
    d = tmp_path / "clip_gelu"
    hf.save_pretrained(d, safe_serialization=True)
    m = jimm_amd.CLIP.from_pretrained(str(d)).eval()
    assert m.vision_model.encoder.layers[0].act == "gelu"
    img = torch.randn(2, 3, 32, 32)
    ids = torch.randint(0, 49, (2, 6))
    ids[:, -1] = 49
    with torch.no_grad():
        ref = hf(pixel_values=img, input_ids=ids).logits_per_image
        out, _ = m(img, ids)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()
