"""HuggingFace <-> hetu_amd checkpoint conversion (Llama family).

Reference parity: python/hetu/models/utils/converter/convert_llama_hf_to_ht.py
(+ convert_utils.py) — maps HF `model.layers.N.self_attn.{q,k,v}_proj` etc.
onto the framework's fused [q|k|v] / [gate|up] column-parallel layout and
writes the sharded-safetensors format utils/checkpoint.py reads.  Both
frameworks use NeoX half-rotation RoPE, so q/k rows transfer without
permutation.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Tuple

import torch


def llama_config_from_hf(hf_cfg: Dict):
    from ..models.llama import LlamaConfig
    return LlamaConfig(
        vocab=hf_cfg["vocab_size"],
        hidden=hf_cfg["hidden_size"],
        n_layer=hf_cfg["num_hidden_layers"],
        n_head=hf_cfg["num_attention_heads"],
        n_kv_head=hf_cfg.get("num_key_value_heads",
                             hf_cfg["num_attention_heads"]),
        ffn_hidden=hf_cfg["intermediate_size"],
        max_seq=hf_cfg.get("max_position_embeddings", 4096),
        rms_eps=hf_cfg.get("rms_norm_eps", 1e-5),
        rope_theta=hf_cfg.get("rope_theta", 10000.0),
    )


def hf_to_hetu_llama(hf: Dict[str, torch.Tensor]
                     ) -> Dict[str, torch.Tensor]:
    """HF Llama state dict -> hetu_amd global names (fused qkv / gate-up)."""
    out: Dict[str, torch.Tensor] = {}
    out["wte.weight"] = hf["model.embed_tokens.weight"]
    out["lnf.weight"] = hf["model.norm.weight"]
    out["lm_head.weight"] = hf.get("lm_head.weight",
                                   hf["model.embed_tokens.weight"])
    i = 0
    while f"model.layers.{i}.self_attn.q_proj.weight" in hf:
        p = f"model.layers.{i}"
        out[f"l{i}.ln1.weight"] = hf[f"{p}.input_layernorm.weight"]
        out[f"l{i}.ln2.weight"] = hf[f"{p}.post_attention_layernorm.weight"]
        out[f"l{i}.attn.wqkv.weight"] = torch.cat(
            [hf[f"{p}.self_attn.q_proj.weight"],
             hf[f"{p}.self_attn.k_proj.weight"],
             hf[f"{p}.self_attn.v_proj.weight"]], dim=0)
        out[f"l{i}.attn.wo.weight"] = hf[f"{p}.self_attn.o_proj.weight"]
        out[f"l{i}.mlp.w_in.weight"] = torch.cat(
            [hf[f"{p}.mlp.gate_proj.weight"],
             hf[f"{p}.mlp.up_proj.weight"]], dim=0)
        out[f"l{i}.mlp.w_out.weight"] = hf[f"{p}.mlp.down_proj.weight"]
        i += 1
    return out


def hetu_to_hf_llama(ht: Dict[str, torch.Tensor], n_head: int,
                     n_kv_head: int, head_dim: int
                     ) -> Dict[str, torch.Tensor]:
    """Inverse mapping: split fused weights back to HF names."""
    out: Dict[str, torch.Tensor] = {}
    out["model.embed_tokens.weight"] = ht["wte.weight"]
    out["model.norm.weight"] = ht["lnf.weight"]
    out["lm_head.weight"] = ht["lm_head.weight"]
    i = 0
    while f"l{i}.attn.wqkv.weight" in ht:
        p = f"model.layers.{i}"
        out[f"{p}.input_layernorm.weight"] = ht[f"l{i}.ln1.weight"]
        out[f"{p}.post_attention_layernorm.weight"] = ht[f"l{i}.ln2.weight"]
        qkv = ht[f"l{i}.attn.wqkv.weight"]
        q, k, v = qkv.split([n_head * head_dim, n_kv_head * head_dim,
                             n_kv_head * head_dim], dim=0)
        out[f"{p}.self_attn.q_proj.weight"] = q
        out[f"{p}.self_attn.k_proj.weight"] = k
        out[f"{p}.self_attn.v_proj.weight"] = v
        out[f"{p}.self_attn.o_proj.weight"] = ht[f"l{i}.attn.wo.weight"]
        w_in = ht[f"l{i}.mlp.w_in.weight"]
        gate, up = w_in.chunk(2, dim=0)
        out[f"{p}.mlp.gate_proj.weight"] = gate
        out[f"{p}.mlp.up_proj.weight"] = up
        out[f"{p}.mlp.down_proj.weight"] = ht[f"l{i}.mlp.w_out.weight"]
        i += 1
    return out


def _read_hf_dir(src: str) -> Tuple[Dict, Dict[str, torch.Tensor]]:
    """Load config.json + all weights from an HF model directory
    (safetensors preferred, .bin fallback)."""
    with open(os.path.join(src, "config.json")) as fh:
        cfg = json.load(fh)
    state: Dict[str, torch.Tensor] = {}
    st_files = sorted(f for f in os.listdir(src)
                      if f.endswith(".safetensors"))
    if st_files:
        from safetensors import safe_open
        for fn in st_files:
            with safe_open(os.path.join(src, fn), framework="pt") as f:
                for k in f.keys():
                    state[k] = f.get_tensor(k)
    else:
        for fn in sorted(f for f in os.listdir(src) if f.endswith(".bin")):
            state.update(torch.load(os.path.join(src, fn),
                                    map_location="cpu", weights_only=True))
    return cfg, state


def convert_llama_hf_to_hetu(src: str, dst: str) -> None:
    """Directory-level conversion: HF Llama checkpoint dir -> hetu_amd
    sharded-safetensors dir readable by utils.checkpoint.load_model."""
    from safetensors.torch import save_file
    cfg, hf_state = _read_hf_dir(src)
    ht_state = hf_to_hetu_llama(hf_state)
    os.makedirs(dst, exist_ok=True)
    fname = "model-r000-00000.safetensors"
    save_file({k: v.contiguous() for k, v in ht_state.items()},
              os.path.join(dst, fname))
    index = {"metadata": {}, "weight_map": {k: fname for k in ht_state}}
    with open(os.path.join(dst, "model.safetensors.index.json"), "w") as fh:
        json.dump(index, fh, indent=1)
    with open(os.path.join(dst, "config.json"), "w") as fh:
        json.dump(cfg, fh, indent=1)


def convert_llama_hetu_to_hf(src: str, dst: str, n_head: int,
                             n_kv_head: int, head_dim: int) -> None:
    from safetensors import safe_open
    from safetensors.torch import save_file
    with open(os.path.join(src, "model.safetensors.index.json")) as fh:
        index = json.load(fh)["weight_map"]
    state: Dict[str, torch.Tensor] = {}
    for fn in sorted(set(index.values())):
        with safe_open(os.path.join(src, fn), framework="pt") as f:
            for k in f.keys():
                state[k] = f.get_tensor(k)
    hf_state = hetu_to_hf_llama(state, n_head, n_kv_head, head_dim)
    os.makedirs(dst, exist_ok=True)
    fname = "model.safetensors"
    save_file({k: v.contiguous() for k, v in hf_state.items()},
              os.path.join(dst, fname))
    index = {"metadata": {}, "weight_map": {k: fname for k in hf_state}}
    with open(os.path.join(dst, "model.safetensors.index.json"), "w") as fh:
        json.dump(index, fh, indent=1)
