"""HF checkpoint interop for CLIP.

Reference mapping: /root/reference/src/jimm/models/clip.py:190-416.
Native layout differences: fused qkv (3H,H); pos-embs stored (1,L,H);
class_embedding stored (1,1,H) (reference reshapes at clip.py:358-361).
"""

from __future__ import annotations

import torch

from jimm_amd.interop.loader import KeyMap, load_checkpoint, save_checkpoint


def _map_tower_layers(m: KeyMap, hf_prefix: str, native_prefix: str, num_layers: int) -> None:
    for i in range(num_layers):
        hf = f"{hf_prefix}.encoder.layers.{i}"
        t = f"{native_prefix}.encoder.layers.{i}"
        for suffix in ("weight", "bias"):
            m.put(
                f"{t}.qkv.{suffix}",
                torch.cat([m.take(f"{hf}.self_attn.{x}_proj.{suffix}") for x in "qkv"], dim=0),
            )
            m.copy(f"{t}.proj.{suffix}", f"{hf}.self_attn.out_proj.{suffix}")
            m.copy(f"{t}.norm1.{suffix}", f"{hf}.layer_norm1.{suffix}")
            m.copy(f"{t}.norm2.{suffix}", f"{hf}.layer_norm2.{suffix}")
            m.copy(f"{t}.fc1.{suffix}", f"{hf}.mlp.fc1.{suffix}")
            m.copy(f"{t}.fc2.{suffix}", f"{hf}.mlp.fc2.{suffix}")


def _parse_config(cfg: dict) -> dict:
    tc, vc = cfg["text_config"], cfg["vision_config"]
    act = (tc.get("hidden_act") or "quick_gelu")
    hidden_act = {"quick_gelu": "quickgelu", "gelu": "gelu",
                  "gelu_new": "gelu_tanh", "gelu_pytorch_tanh": "gelu_tanh"}.get(act, "quickgelu")
    return dict(
        embed_dim=cfg.get("projection_dim", 512),
        hidden_act=hidden_act,
        image_resolution=vc.get("image_size", 224),
        vision_layers=vc["num_hidden_layers"],
        vision_width=vc["hidden_size"],
        vision_patch_size=vc.get("patch_size", 32),
        context_length=tc.get("max_position_embeddings", 77),
        vocab_size=tc.get("vocab_size", 49408),
        transformer_width=tc["hidden_size"],
        transformer_heads=tc["num_attention_heads"],
        transformer_layers=tc["num_hidden_layers"],
        vision_mlp_dim=vc.get("intermediate_size"),
        transformer_mlp_dim=tc.get("intermediate_size"),
    )


def _infer_config(sd: dict[str, torch.Tensor]) -> dict:
    """Shape inference from a bare checkpoint (reference clip.py:208-247)."""
    vision_width = sd["vision_model.embeddings.class_embedding"].shape[-1]
    conv_w = sd["vision_model.embeddings.patch_embedding.weight"]
    patch = conv_w.shape[-1]
    import math

    v_pos = sd["vision_model.embeddings.position_embedding.weight"].shape[0]
    img = int(math.isqrt(v_pos - 1)) * patch
    n_layers = lambda pref: 1 + max(int(k.split(".")[3]) for k in sd if k.startswith(pref + ".encoder.layers."))
    tw = sd["text_model.embeddings.token_embedding.weight"].shape[1]
    return dict(
        embed_dim=sd["visual_projection.weight"].shape[0],
        image_resolution=img,
        vision_layers=n_layers("vision_model"),
        vision_width=vision_width,
        vision_patch_size=patch,
        context_length=sd["text_model.embeddings.position_embedding.weight"].shape[0],
        vocab_size=sd["text_model.embeddings.token_embedding.weight"].shape[0],
        transformer_width=tw,
        transformer_heads=max(1, tw // 64),  # clip.py:368
        transformer_layers=n_layers("text_model"),
        vision_mlp_dim=sd["vision_model.encoder.layers.0.mlp.fc1.weight"].shape[0],
        transformer_mlp_dim=sd["text_model.encoder.layers.0.mlp.fc1.weight"].shape[0],
    )


def map_clip(sd: dict[str, torch.Tensor], vision_layers: int, text_layers: int) -> KeyMap:
    m = KeyMap(sd)
    ls = m.take("logit_scale")
    m.put("logit_scale", ls.reshape(()))
    m.copy("visual_projection.weight", "visual_projection.weight")
    m.copy("text_projection.weight", "text_projection.weight")
    # vision tower
    m.put("vision_model.cls_token", m.take("vision_model.embeddings.class_embedding").reshape(1, 1, -1))
    m.copy("vision_model.patch_weight", "vision_model.embeddings.patch_embedding.weight")
    m.put("vision_model.pos_embedding", m.take("vision_model.embeddings.position_embedding.weight").unsqueeze(0))
    m.copy("vision_model.ln_pre.weight", "vision_model.pre_layrnorm.weight")  # HF's typo'd name
    m.copy("vision_model.ln_pre.bias", "vision_model.pre_layrnorm.bias")
    m.copy("vision_model.ln_post.weight", "vision_model.post_layernorm.weight")
    m.copy("vision_model.ln_post.bias", "vision_model.post_layernorm.bias")
    _map_tower_layers(m, "vision_model", "vision_model", vision_layers)
    # text tower
    m.copy("text_model.token_embedding.weight", "text_model.embeddings.token_embedding.weight")
    m.put("text_model.pos_embedding", m.take("text_model.embeddings.position_embedding.weight").unsqueeze(0))
    m.copy("text_model.ln_final.weight", "text_model.final_layer_norm.weight")
    m.copy("text_model.ln_final.bias", "text_model.final_layer_norm.bias")
    _map_tower_layers(m, "text_model", "text_model", text_layers)
    return m


def load_clip(cls, model_name_or_path: str, *, use_pytorch: bool = False, dtype: torch.dtype = torch.float32, device="cpu"):
    sd, cfg = load_checkpoint(model_name_or_path, use_pytorch=use_pytorch)
    kwargs = _parse_config(cfg) if cfg else _infer_config(sd)
    model = cls(**kwargs)
    m = map_clip(sd, len(model.vision_model.encoder.layers), len(model.text_model.encoder.layers))
    m.finish(model, dtype=dtype)
    return model.to(device)


def _unmap_tower_layers(sd, out, hf_prefix: str, native_prefix: str, num_layers: int) -> None:
    for i in range(num_layers):
        hf = f"{hf_prefix}.encoder.layers.{i}"
        t = f"{native_prefix}.encoder.layers.{i}"
        for suffix in ("weight", "bias"):
            q, k, v = sd[f"{t}.qkv.{suffix}"].chunk(3, dim=0)
            out[f"{hf}.self_attn.q_proj.{suffix}"] = q
            out[f"{hf}.self_attn.k_proj.{suffix}"] = k
            out[f"{hf}.self_attn.v_proj.{suffix}"] = v
            out[f"{hf}.self_attn.out_proj.{suffix}"] = sd[f"{t}.proj.{suffix}"]
            out[f"{hf}.layer_norm1.{suffix}"] = sd[f"{t}.norm1.{suffix}"]
            out[f"{hf}.layer_norm2.{suffix}"] = sd[f"{t}.norm2.{suffix}"]
            out[f"{hf}.mlp.fc1.{suffix}"] = sd[f"{t}.fc1.{suffix}"]
            out[f"{hf}.mlp.fc2.{suffix}"] = sd[f"{t}.fc2.{suffix}"]


def save_clip(model, save_dir: str) -> None:
    sd = model.state_dict()
    out: dict[str, torch.Tensor] = {}
    out["logit_scale"] = sd["logit_scale"]
    out["visual_projection.weight"] = sd["visual_projection.weight"]
    out["text_projection.weight"] = sd["text_projection.weight"]
    out["vision_model.embeddings.class_embedding"] = sd["vision_model.cls_token"].reshape(-1)
    out["vision_model.embeddings.patch_embedding.weight"] = sd["vision_model.patch_weight"]
    out["vision_model.embeddings.position_embedding.weight"] = sd["vision_model.pos_embedding"].squeeze(0)
    out["vision_model.pre_layrnorm.weight"] = sd["vision_model.ln_pre.weight"]
    out["vision_model.pre_layrnorm.bias"] = sd["vision_model.ln_pre.bias"]
    out["vision_model.post_layernorm.weight"] = sd["vision_model.ln_post.weight"]
    out["vision_model.post_layernorm.bias"] = sd["vision_model.ln_post.bias"]
    out["text_model.embeddings.token_embedding.weight"] = sd["text_model.token_embedding.weight"]
    out["text_model.embeddings.position_embedding.weight"] = sd["text_model.pos_embedding"].squeeze(0)
    out["text_model.final_layer_norm.weight"] = sd["text_model.ln_final.weight"]
    out["text_model.final_layer_norm.bias"] = sd["text_model.ln_final.bias"]
    nv, nt = len(model.vision_model.encoder.layers), len(model.text_model.encoder.layers)
    _unmap_tower_layers(sd, out, "vision_model", "vision_model", nv)
    _unmap_tower_layers(sd, out, "text_model", "text_model", nt)
    vb, tb = model.vision_model, model.text_model
    cfg = {
        "model_type": "clip",
        "architectures": ["CLIPModel"],
        "projection_dim": model.embed_dim,
        "text_config": {
            "model_type": "clip_text_model",
            "hidden_size": tb.token_embedding.embedding_dim,
            "intermediate_size": tb.encoder.layers[0].fc1.out_features,
            "num_hidden_layers": nt,
            "num_attention_heads": tb.encoder.layers[0].num_heads,
            "max_position_embeddings": tb.pos_embedding.shape[1],
            "vocab_size": tb.token_embedding.num_embeddings,
            "hidden_act": "quick_gelu",
        },
        "vision_config": {
            "model_type": "clip_vision_model",
            "hidden_size": vb.hidden_size,
            "intermediate_size": vb.encoder.layers[0].fc1.out_features,
            "num_hidden_layers": nv,
            "num_attention_heads": vb.encoder.layers[0].num_heads,
            "image_size": vb.img_size,
            "patch_size": vb.patch_size,
            "hidden_act": "quick_gelu",
        },
    }
    save_checkpoint(out, cfg, save_dir)
