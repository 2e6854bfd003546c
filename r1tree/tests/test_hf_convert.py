"""HF Llama checkpoint conversion: logits must match transformers.

Reference parity: python/hetu/models/utils/converter/convert_llama_hf_to_ht.py
— a converted HF checkpoint loaded into the framework must produce the same
forward as the HF implementation.
"""
import json
import os

import pytest
import torch

transformers = pytest.importorskip("transformers")


def _tiny_hf(tmp_path):
    from transformers import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig(vocab_size=257, hidden_size=64,
                      intermediate_size=112, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=4,
                      max_position_embeddings=64, rms_norm_eps=1e-6,
                      rope_theta=10000.0, tie_word_embeddings=False,
                      attn_implementation="eager")
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg).eval().float()
    src = str(tmp_path / "hf")
    model.save_pretrained(src, safe_serialization=True)
    return model, src


def test_llama_hf_roundtrip_logits(tmp_path):
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.models.llama import build_llama_train_graph
    from hetu_amd.utils.checkpoint import load_model
    from hetu_amd.utils.hf_convert import (convert_llama_hetu_to_hf,
                                           convert_llama_hf_to_hetu,
                                           llama_config_from_hf)

    hf_model, src = _tiny_hf(tmp_path)
    dst = str(tmp_path / "ht")
    convert_llama_hf_to_hetu(src, dst)

    with open(os.path.join(src, "config.json")) as fh:
        cfg = llama_config_from_hf(json.load(fh))
    assert cfg.hidden == 64 and cfg.n_layer == 2 and cfg.ffn_hidden == 112

    B, S = 2, 16
    g, h = build_llama_train_graph(cfg, B, S, dtype=torch.float32)
    missing = load_model(g.parameters, dst, comm=None)
    assert not missing
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    torch.manual_seed(3)
    ids = torch.randint(0, cfg.vocab, (B, S))
    labels = torch.randint(0, cfg.vocab, (B * S,))
    logits, = g.run([h["logits"]], {h["input_ids"]: ids,
                                    h["labels"]: labels}, ctx=ctx)
    with torch.no_grad():
        ref = hf_model(ids).logits
    got = logits.reshape(B, S, cfg.vocab).float()
    err = (got - ref).abs().max().item()
    assert err < 2e-3, err

    # round trip back to HF layout and compare raw tensors
    back = str(tmp_path / "hf2")
    convert_llama_hetu_to_hf(dst, back, cfg.n_head, cfg.n_kv_head,
                             cfg.hidden // cfg.n_head)
    from safetensors import safe_open
    sd = hf_model.state_dict()
    with safe_open(os.path.join(back, "model.safetensors"),
                   framework="pt") as f:
        for k in f.keys():
            assert torch.equal(f.get_tensor(k), sd[k]), k
