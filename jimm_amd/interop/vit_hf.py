"""HF checkpoint interop for ViT.

Reference mapping: /root/reference/src/jimm/models/vit.py:105-273.
Differences from the reference's transforms (vit.py:239-250): our native
layout is torch-convention (out,in) with a FUSED qkv (3H,H), so q/k/v are
concatenated instead of reshaped to (H, heads, head_dim).

Supports BOTH HF key schemes:
  * classic hub naming (transformers <=4.x, what real hub checkpoints use):
    ``vit.encoder.layer.N.attention.attention.query.weight`` ...
  * transformers 5.x naming: ``vit.layers.N.attention.q_proj.weight`` ...
"""

from __future__ import annotations

import math

import torch

from jimm_amd.interop.loader import KeyMap, load_checkpoint, save_checkpoint


def _parse_config(cfg: dict) -> dict:
    """HF config.json -> constructor kwargs (reference vit.py:131-142)."""
    act = cfg.get("hidden_act", "gelu")
    hidden_act = {"gelu": "gelu", "gelu_new": "gelu_tanh", "gelu_pytorch_tanh": "gelu_tanh", "quick_gelu": "quickgelu"}.get(act, "gelu")
    id2label = cfg.get("id2label") or {}
    return dict(
        hidden_size=cfg["hidden_size"],
        num_layers=cfg["num_hidden_layers"],
        num_heads=cfg["num_attention_heads"],
        mlp_dim=cfg["intermediate_size"],
        img_size=cfg.get("image_size", 224),
        patch_size=cfg.get("patch_size", 16),
        num_classes=len(id2label) if id2label else 1000,
        hidden_act=hidden_act,
        layernorm_epsilon=cfg.get("layer_norm_eps", 1e-12),
    )


def _infer_config(sd: dict[str, torch.Tensor]) -> dict:
    """Shape inference from a bare checkpoint (reference vit.py:144-164)."""
    hidden = sd["vit.embeddings.cls_token"].shape[-1]
    layer_ids = set()
    for k in sd:
        for prefix in ("vit.encoder.layer.", "vit.layers."):
            if k.startswith(prefix):
                layer_ids.add(int(k[len(prefix):].split(".")[0]))
    conv_w = sd["vit.embeddings.patch_embeddings.projection.weight"]
    patch = conv_w.shape[-1]
    pos_len = sd["vit.embeddings.position_embeddings"].shape[1]
    img = int(math.isqrt(pos_len - 1)) * patch
    fc1_key = next(k for k in sd if k.endswith(("intermediate.dense.weight", "mlp.fc1.weight")))
    has_cls = "classifier.weight" in sd
    return dict(
        hidden_size=hidden,
        num_layers=max(layer_ids) + 1,
        num_heads=max(1, hidden // 64),  # assumed head_dim 64 (vit.py:156-157)
        mlp_dim=sd[fc1_key].shape[0],
        img_size=img,
        patch_size=patch,
        num_classes=sd["classifier.weight"].shape[0] if has_cls else 1000,
        do_classification=has_cls,
    )


def _layer_keys(sd: dict, i: int) -> dict[str, str]:
    """Resolve per-layer HF key names for either naming scheme."""
    old = f"vit.encoder.layer.{i}"
    new = f"vit.layers.{i}"
    if f"{old}.attention.attention.query.weight" in sd:
        return {
            "q": f"{old}.attention.attention.query",
            "k": f"{old}.attention.attention.key",
            "v": f"{old}.attention.attention.value",
            "o": f"{old}.attention.output.dense",
            "fc1": f"{old}.intermediate.dense",
            "fc2": f"{old}.output.dense",
            "ln1": f"{old}.layernorm_before",
            "ln2": f"{old}.layernorm_after",
        }
    return {
        "q": f"{new}.attention.q_proj",
        "k": f"{new}.attention.k_proj",
        "v": f"{new}.attention.v_proj",
        "o": f"{new}.attention.o_proj",
        "fc1": f"{new}.mlp.fc1",
        "fc2": f"{new}.mlp.fc2",
        "ln1": f"{new}.layernorm_before",
        "ln2": f"{new}.layernorm_after",
    }


def map_vit(sd: dict[str, torch.Tensor], num_layers: int, do_classification: bool) -> KeyMap:
    m = KeyMap(sd)
    m.copy("vision.cls_token", "vit.embeddings.cls_token")
    m.copy("vision.pos_embedding", "vit.embeddings.position_embeddings")
    m.copy("vision.patch_weight", "vit.embeddings.patch_embeddings.projection.weight")
    m.copy("vision.patch_bias", "vit.embeddings.patch_embeddings.projection.bias")
    m.copy("vision.ln_post.weight", "vit.layernorm.weight")
    m.copy("vision.ln_post.bias", "vit.layernorm.bias")
    if do_classification:
        m.copy("classifier.weight", "classifier.weight")
        m.copy("classifier.bias", "classifier.bias")
    for i in range(num_layers):
        hf = _layer_keys(sd, i)
        t = f"vision.encoder.layers.{i}"
        for suffix in ("weight", "bias"):
            m.put(f"{t}.qkv.{suffix}", torch.cat([m.take(f"{hf[x]}.{suffix}") for x in "qkv"], dim=0))
            m.copy(f"{t}.proj.{suffix}", f"{hf['o']}.{suffix}")
            m.copy(f"{t}.fc1.{suffix}", f"{hf['fc1']}.{suffix}")
            m.copy(f"{t}.fc2.{suffix}", f"{hf['fc2']}.{suffix}")
            m.copy(f"{t}.norm1.{suffix}", f"{hf['ln1']}.{suffix}")
            m.copy(f"{t}.norm2.{suffix}", f"{hf['ln2']}.{suffix}")
    return m


def load_vit(cls, model_name_or_path: str, *, use_pytorch: bool = False, dtype: torch.dtype = torch.float32, device="cpu"):
    sd, cfg = load_checkpoint(model_name_or_path, use_pytorch=use_pytorch)
    kwargs = _parse_config(cfg) if cfg else _infer_config(sd)
    if cfg is not None:
        kwargs["do_classification"] = "classifier.weight" in sd
        if kwargs["do_classification"]:
            kwargs["num_classes"] = sd["classifier.weight"].shape[0]
    model = cls(**kwargs)
    m = map_vit(sd, len(model.vision.encoder.layers), model.do_classification)
    m.finish(model, dtype=dtype)
    return model.to(device)


def save_vit(model, save_dir: str) -> None:
    """Inverse mapping -> classic hub key names + config.json."""
    sd = model.state_dict()
    out: dict[str, torch.Tensor] = {}
    out["vit.embeddings.cls_token"] = sd["vision.cls_token"]
    out["vit.embeddings.position_embeddings"] = sd["vision.pos_embedding"]
    out["vit.embeddings.patch_embeddings.projection.weight"] = sd["vision.patch_weight"]
    out["vit.embeddings.patch_embeddings.projection.bias"] = sd["vision.patch_bias"]
    out["vit.layernorm.weight"] = sd["vision.ln_post.weight"]
    out["vit.layernorm.bias"] = sd["vision.ln_post.bias"]
    if model.do_classification:
        out["classifier.weight"] = sd["classifier.weight"]
        out["classifier.bias"] = sd["classifier.bias"]
    for i in range(len(model.vision.encoder.layers)):
        t = f"vision.encoder.layers.{i}"
        hf = f"vit.encoder.layer.{i}"
        for suffix in ("weight", "bias"):
            q, k, v = sd[f"{t}.qkv.{suffix}"].chunk(3, dim=0)
            out[f"{hf}.attention.attention.query.{suffix}"] = q
            out[f"{hf}.attention.attention.key.{suffix}"] = k
            out[f"{hf}.attention.attention.value.{suffix}"] = v
            out[f"{hf}.attention.output.dense.{suffix}"] = sd[f"{t}.proj.{suffix}"]
            out[f"{hf}.intermediate.dense.{suffix}"] = sd[f"{t}.fc1.{suffix}"]
            out[f"{hf}.output.dense.{suffix}"] = sd[f"{t}.fc2.{suffix}"]
            out[f"{hf}.layernorm_before.{suffix}"] = sd[f"{t}.norm1.{suffix}"]
            out[f"{hf}.layernorm_after.{suffix}"] = sd[f"{t}.norm2.{suffix}"]
    vb = model.vision
    blk = vb.encoder.layers[0]
    cfg = {
        "model_type": "vit",
        "architectures": ["ViTForImageClassification"],
        "hidden_size": vb.hidden_size,
        "num_hidden_layers": len(vb.encoder.layers),
        "num_attention_heads": blk.num_heads,
        "intermediate_size": blk.fc1.out_features,
        "image_size": vb.img_size,
        "patch_size": vb.patch_size,
        "hidden_act": {"gelu": "gelu", "gelu_tanh": "gelu_pytorch_tanh", "quickgelu": "quick_gelu"}[blk.act],
        "layer_norm_eps": vb.eps,
        "id2label": {str(i): f"LABEL_{i}" for i in range(model.classifier.out_features)} if model.do_classification else {},
    }
    save_checkpoint(out, cfg, save_dir)
