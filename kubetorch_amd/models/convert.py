"""HF-checkpoint conversion for the Llama family.

Users arriving from the reference bring HuggingFace-format Llama weights;
this maps them onto kubetorch_amd's fused layout:

    q_proj + k_proj + v_proj  ->  attn.wqkv   (one GEMM per layer)
    gate_proj + up_proj       ->  mlp.w_gate_up
    o_proj -> attn.wo, down_proj -> mlp.w_down
    input_layernorm -> attn_norm, post_attention_layernorm -> mlp_norm

plus the RoPE convention change: HF stores q/k with interleaved rotary
pairs permuted into block-half order per head — the SAME rotate-half
convention this model uses, so weights map 1:1 (HF's "permute" already
happened at HF-conversion time for Llama).
"""
import json
import os

import torch


def _hf_layer_map(i):
    p = f"model.layers.{i}."
    return {
        "wqkv": [p + "self_attn.q_proj.weight", p + "self_attn.k_proj.weight",
                 p + "self_attn.v_proj.weight"],
        "wo": p + "self_attn.o_proj.weight",
        "w_gate_up": [p + "mlp.gate_proj.weight", p + "mlp.up_proj.weight"],
        "w_down": p + "mlp.down_proj.weight",
        "attn_norm": p + "input_layernorm.weight",
        "mlp_norm": p + "post_attention_layernorm.weight",
    }


def hf_to_kt_state_dict(hf_sd, cfg):
    """Map an HF Llama state dict onto the kubetorch_amd Llama layout
    (concatenating the fused projections). Returns a new state dict."""
    out = {}
    out["embed.weight"] = hf_sd["model.embed_tokens.weight"]
    out["norm.weight"] = hf_sd["model.norm.weight"]
    if "lm_head.weight" in hf_sd:
        out["lm_head.weight"] = hf_sd["lm_head.weight"]
    else:  # tied embeddings
        out["lm_head.weight"] = hf_sd["model.embed_tokens.weight"]
    for i in range(cfg.n_layers):
        m = _hf_layer_map(i)
        q, k, v = (hf_sd[n] for n in m["wqkv"])
        out[f"layers.{i}.attn.wqkv.weight"] = torch.cat([q, k, v], dim=0)
        out[f"layers.{i}.attn.wo.weight"] = hf_sd[m["wo"]]
        g, u = (hf_sd[n] for n in m["w_gate_up"])
        out[f"layers.{i}.mlp.w_gate_up.weight"] = torch.cat([g, u], dim=0)
        out[f"layers.{i}.mlp.w_down.weight"] = hf_sd[m["w_down"]]
        out[f"layers.{i}.attn_norm.weight"] = hf_sd[m["attn_norm"]]
        out[f"layers.{i}.mlp_norm.weight"] = hf_sd[m["mlp_norm"]]
    return out


def kt_to_hf_state_dict(kt_sd, cfg):
    """Inverse mapping (export a trained model back to HF format)."""
    d, hd = cfg.dim, cfg.head_dim
    nq, nkv = cfg.n_heads * hd, cfg.n_kv_heads * hd
    out = {
        "model.embed_tokens.weight": kt_sd["embed.weight"],
        "model.norm.weight": kt_sd["norm.weight"],
        "lm_head.weight": kt_sd["lm_head.weight"],
    }
    for i in range(cfg.n_layers):
        m = _hf_layer_map(i)
        qkv = kt_sd[f"layers.{i}.attn.wqkv.weight"]
        out[m["wqkv"][0]] = qkv[:nq]
        out[m["wqkv"][1]] = qkv[nq:nq + nkv]
        out[m["wqkv"][2]] = qkv[nq + nkv:]
        out[m["wo"]] = kt_sd[f"layers.{i}.attn.wo.weight"]
        gu = kt_sd[f"layers.{i}.mlp.w_gate_up.weight"]
        out[m["w_gate_up"][0]] = gu[:cfg.intermediate]
        out[m["w_gate_up"][1]] = gu[cfg.intermediate:]
        out[m["w_down"]] = kt_sd[f"layers.{i}.mlp.w_down.weight"]
        out[m["attn_norm"]] = kt_sd[f"layers.{i}.attn_norm.weight"]
        out[m["mlp_norm"]] = kt_sd[f"layers.{i}.mlp_norm.weight"]
    return out


def config_from_hf(path):
    """Build a LlamaConfig from an HF config.json."""
    from kubetorch_amd.models.llama import LlamaConfig

    with open(os.path.join(path, "config.json")) as f:
        hc = json.load(f)
    return LlamaConfig(
        dim=hc["hidden_size"],
        n_layers=hc["num_hidden_layers"],
        n_heads=hc["num_attention_heads"],
        n_kv_heads=hc.get("num_key_value_heads", hc["num_attention_heads"]),
        intermediate=hc["intermediate_size"],
        vocab_size=hc["vocab_size"],
        max_seq_len=hc.get("max_position_embeddings", 8192),
        rope_base=hc.get("rope_theta", 500000.0),
        norm_eps=hc.get("rms_norm_eps", 1e-5),
    )


def load_hf_checkpoint(path, device="cpu", dtype=torch.bfloat16):
    """Load an HF Llama checkpoint directory (config.json +
    *.safetensors shards) into a kubetorch_amd Llama. Returns the model."""
    from safetensors.torch import load_file

    from kubetorch_amd.models.llama import Llama

    cfg = config_from_hf(path)
    hf_sd = {}
    shards = sorted(f for f in os.listdir(path) if f.endswith(".safetensors"))
    if not shards:
        raise FileNotFoundError(f"no .safetensors shards under {path}")
    for shard in shards:
        hf_sd.update(load_file(os.path.join(path, shard), device="cpu"))
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
        with torch.device("meta"):
            model = Llama(cfg)
        model = model.to_empty(device=device)
        # rope tables were materialized empty by to_empty: recompute
        from kubetorch_amd import ops

        cos, sin = ops.precompute_rope(cfg.max_seq_len, cfg.head_dim,
                                       cfg.rope_base, device=device)
        model.rope_cos, model.rope_sin = cos, sin
    finally:
        torch.set_default_dtype(prev)
    sd = hf_to_kt_state_dict(hf_sd, cfg)
    sd = {k: v.to(device=device, dtype=dtype) for k, v in sd.items()}
    missing, unexpected = model.load_state_dict(sd, strict=False)
    missing = [m for m in missing if not m.startswith("rope_")]
    if missing or unexpected:
        raise KeyError(f"HF mapping mismatch: missing={missing[:5]} "
                       f"unexpected={unexpected[:5]}")
    return model
