"""Inference/serving path: KV-cached autoregressive generation.

Reference context: Hetu's user-facing model stack is training-first; for
serving parity this engine runs the same Llama/GPT weights with a
fixed-capacity KV cache.  MI355X-native: prefill goes through the
hand-written flash-attention kernel (one pass over the prompt), decode
steps use cached K/V with a single fused read per layer; all norms /
RoPE / SwiGLU hit the same HIP kernels as training via ops.functional.

Weights load from a hetu_amd checkpoint dir (utils/checkpoint format,
incl. checkpoints converted from HF with utils/hf_convert) or from an
in-memory state dict.
"""
from __future__ import annotations

import math
from typing import Dict, Optional

import torch

from ..models.llama import LlamaConfig, rope_tables
from ..ops import functional as F


def sample_next(logits: torch.Tensor, temperature: float = 0.0,
                top_k: int = 0, top_p: float = 0.0,
                gen: Optional[torch.Generator] = None) -> torch.Tensor:
    """Next-token sampling: greedy (temperature<=0), temperature,
    top-k, and nucleus (top-p) filters compose.  CPU multinomial with an
    explicit generator keeps runs reproducible across devices."""
    if temperature <= 0:
        return logits.argmax(-1)
    lg = logits.float() / temperature
    if top_k:
        kth = lg.topk(top_k, dim=-1).values[:, -1:]
        lg = lg.masked_fill(lg < kth, float("-inf"))
    if top_p and 0.0 < top_p < 1.0:
        srt, idx = lg.sort(dim=-1, descending=True)
        cum = torch.softmax(srt, dim=-1).cumsum(dim=-1)
        # keep the smallest prefix with cumulative prob >= top_p (the
        # first token always survives)
        drop_sorted = cum - torch.softmax(srt, dim=-1) >= top_p
        drop = drop_sorted.scatter(-1, idx, drop_sorted)
        lg = lg.masked_fill(drop, float("-inf"))
    probs = torch.softmax(lg, dim=-1).cpu()
    return torch.multinomial(probs, 1, generator=gen).squeeze(-1) \
        .to(logits.device)


class LlamaKVCache:
    def __init__(self, cfg: LlamaConfig, batch: int, max_len: int,
                 device, dtype):
        dh = cfg.hidden // cfg.n_head
        shape = (cfg.n_layer, batch, cfg.n_kv_head, max_len, dh)
        self.k = torch.zeros(shape, device=device, dtype=dtype)
        self.v = torch.zeros(shape, device=device, dtype=dtype)
        self.len = 0


class LlamaGenerator:
    def __init__(self, cfg: LlamaConfig, state: Dict[str, torch.Tensor],
                 device: Optional[torch.device] = None,
                 dtype: torch.dtype = torch.float32):
        self.cfg = cfg
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.dtype = dtype
        self.w = {k: v.to(self.device, dtype) for k, v in state.items()}
        dh = cfg.hidden // cfg.n_head
        cos, sin = rope_tables(cfg, cfg.max_seq, torch.float32)
        self.cos = cos.to(self.device)
        self.sin = sin.to(self.device)
        self.scale = 1.0 / math.sqrt(dh)

    @classmethod
    def from_checkpoint(cls, cfg: LlamaConfig, path: str, **kw
                        ) -> "LlamaGenerator":
        import json
        import os

        from safetensors import safe_open
        idx = os.path.join(path, "model.safetensors.index.json")
        if os.path.exists(idx):
            with open(idx) as fh:
                index = json.load(fh)["weight_map"]
        else:
            # single-shard HF convention: plain model.safetensors, no index
            index = {"": "model.safetensors"}
        state = {}
        for fn in sorted(set(index.values())):
            with safe_open(os.path.join(path, fn), framework="pt") as f:
                for k in f.keys():
                    state[k] = f.get_tensor(k)
        return cls(cfg, state, **kw)

    # ---- building blocks -------------------------------------------------
    def _attn(self, x, layer, cache: LlamaKVCache, pos0: int):
        cfg = self.cfg
        B, S, _ = x.shape
        H, Hkv = cfg.n_head, cfg.n_kv_head
        dh = cfg.hidden // H
        qkv = x @ self.w[f"l{layer}.attn.wqkv.weight"].t()
        q, k, v = qkv.split([H * dh, Hkv * dh, Hkv * dh], dim=-1)
        q = q.view(B, S, H, dh)
        k = k.view(B, S, Hkv, dh)
        cos = self.cos[pos0:pos0 + S]
        sin = self.sin[pos0:pos0 + S]
        q = F.rope_fwd(q, cos, sin)
        k = F.rope_fwd(k, cos, sin)
        v = v.view(B, S, Hkv, dh)
        # append to cache
        cache.k[layer][:, :, pos0:pos0 + S] = k.permute(0, 2, 1, 3)
        cache.v[layer][:, :, pos0:pos0 + S] = v.permute(0, 2, 1, 3)
        kk = cache.k[layer][:, :, :pos0 + S]
        vv = cache.v[layer][:, :, :pos0 + S]
        qt = q.permute(0, 2, 1, 3)
        if S > 1 and qt.is_cuda and dh == 128 \
                and qt.dtype == torch.bfloat16:
            o, _ = F.flash_attn_fwd(qt.contiguous(), kk.contiguous(),
                                    vv.contiguous(), True, self.scale)
        else:
            rep = H // Hkv
            kr = kk.repeat_interleave(rep, dim=1) if rep > 1 else kk
            vr = vv.repeat_interleave(rep, dim=1) if rep > 1 else vv
            mask = None
            if S > 1:
                Skv = pos0 + S
                mask = torch.ones(S, Skv, dtype=torch.bool,
                                  device=x.device).tril(diagonal=Skv - S)
            scores = (qt.float() @ kr.float().transpose(-1, -2)) \
                * self.scale
            if mask is not None:
                scores = scores.masked_fill(~mask, float("-inf"))
            p = torch.softmax(scores, dim=-1)
            o = (p @ vr.float()).to(x.dtype)
        o = o.permute(0, 2, 1, 3).reshape(B, S, H * dh)
        return o @ self.w[f"l{layer}.attn.wo.weight"].t()

    def _block(self, x, layer, cache, pos0):
        cfg = self.cfg
        h = F.rmsnorm_fwd(x, self.w[f"l{layer}.ln1.weight"],
                          cfg.rms_eps)[0]
        x = x + self._attn(h, layer, cache, pos0)
        h = F.rmsnorm_fwd(x, self.w[f"l{layer}.ln2.weight"],
                          cfg.rms_eps)[0]
        h = h @ self.w[f"l{layer}.mlp.w_in.weight"].t()
        h = F.swiglu_fwd(h)
        x = x + h @ self.w[f"l{layer}.mlp.w_out.weight"].t()
        return x

    def _forward(self, ids, cache, pos0):
        x = self.w["wte.weight"][ids]
        for layer in range(self.cfg.n_layer):
            x = self._block(x, layer, cache, pos0)
        x = F.rmsnorm_fwd(x, self.w["lnf.weight"], self.cfg.rms_eps)[0]
        return x[:, -1] @ self.w["lm_head.weight"].t()   # last-token logits

    # ---- public API ------------------------------------------------------
    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 top_p: float = 0.0, eos_id: Optional[int] = None,
                 seed: int = 0) -> torch.Tensor:
        """input_ids [B, S] -> [B, S + new]; temperature 0 = greedy."""
        ids = input_ids.to(self.device)
        B, S = ids.shape
        cache = LlamaKVCache(self.cfg, B, S + max_new_tokens, self.device,
                             self.dtype)
        gen = torch.Generator(device="cpu").manual_seed(seed)
        logits = self._forward(ids, cache, 0)
        pos = S
        out = [ids]
        alive = torch.ones(B, dtype=torch.bool)
        for _ in range(max_new_tokens):
            nxt = sample_next(logits, temperature, top_k, top_p, gen)
            out.append(nxt.unsqueeze(1))
            if eos_id is not None:
                alive &= (nxt.cpu() != eos_id)
                if not alive.any():
                    break
            logits = self._forward(nxt.unsqueeze(1), cache, pos)
            pos += 1
        return torch.cat(out, dim=1)


class GPTGenerator:
    """KV-cached generation for the GPT family (learned positions,
    LayerNorm, biased linears, gelu MLP)."""

    def __init__(self, cfg, state: Dict[str, torch.Tensor],
                 device: Optional[torch.device] = None,
                 dtype: torch.dtype = torch.float32):
        self.cfg = cfg
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.dtype = dtype
        self.w = {k: v.to(self.device, dtype) for k, v in state.items()}
        self.scale = 1.0 / math.sqrt(cfg.hidden // cfg.n_head)

    def _lin(self, x, name):
        y = x @ self.w[f"{name}.weight"].t()
        b = self.w.get(f"{name}.bias")
        return y + b if b is not None else y

    def _attn(self, x, layer, kcache, vcache, pos0):
        cfg = self.cfg
        B, S, _ = x.shape
        H = cfg.n_head
        dh = cfg.hidden // H
        qkv = self._lin(x, f"h{layer}.attn.wqkv")
        q, k, v = qkv.chunk(3, -1)
        q = q.view(B, S, H, dh).permute(0, 2, 1, 3)
        kcache[layer][:, :, pos0:pos0 + S] = \
            k.view(B, S, H, dh).permute(0, 2, 1, 3)
        vcache[layer][:, :, pos0:pos0 + S] = \
            v.view(B, S, H, dh).permute(0, 2, 1, 3)
        kk = kcache[layer][:, :, :pos0 + S]
        vv = vcache[layer][:, :, :pos0 + S]
        if S > 1 and q.is_cuda and dh == 128 and q.dtype == torch.bfloat16:
            o, _ = F.flash_attn_fwd(q.contiguous(), kk.contiguous(),
                                    vv.contiguous(), True, self.scale)
        else:
            scores = (q.float() @ kk.float().transpose(-1, -2)) * self.scale
            if S > 1:
                Skv = pos0 + S
                mask = torch.ones(S, Skv, dtype=torch.bool,
                                  device=x.device).tril(diagonal=Skv - S)
                scores = scores.masked_fill(~mask, float("-inf"))
            o = (torch.softmax(scores, -1) @ vv.float()).to(x.dtype)
        o = o.permute(0, 2, 1, 3).reshape(B, S, H * dh)
        return self._lin(o, f"h{layer}.attn.wo")

    def _forward(self, ids, kcache, vcache, pos0):
        cfg = self.cfg
        S = ids.shape[1]
        pos = torch.arange(pos0, pos0 + S, device=self.device)
        x = self.w["wte.weight"][ids] + self.w["wpe.weight"][pos]
        for i in range(cfg.n_layer):
            h = F.layernorm_fwd(x, self.w[f"h{i}.ln1.weight"],
                                self.w[f"h{i}.ln1.bias"], 1e-5)[0]
            x = x + self._attn(h, i, kcache, vcache, pos0)
            h = F.layernorm_fwd(x, self.w[f"h{i}.ln2.weight"],
                                self.w[f"h{i}.ln2.bias"], 1e-5)[0]
            h = F.gelu_fwd(self._lin(h, f"h{i}.mlp.wfc"))
            x = x + self._lin(h, f"h{i}.mlp.wproj")
        x = F.layernorm_fwd(x, self.w["lnf.weight"], self.w["lnf.bias"],
                            1e-5)[0]
        return x[:, -1] @ self.w["lm_head.weight"].t()

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 top_p: float = 0.0, seed: int = 0) -> torch.Tensor:
        cfg = self.cfg
        ids = input_ids.to(self.device)
        B, S = ids.shape
        dh = cfg.hidden // cfg.n_head
        L = S + max_new_tokens
        kc = torch.zeros(cfg.n_layer, B, cfg.n_head, L, dh,
                         device=self.device, dtype=self.dtype)
        vc = torch.zeros_like(kc)
        gen = torch.Generator(device="cpu").manual_seed(seed)
        logits = self._forward(ids, kc, vc, 0)
        out = [ids]
        pos = S
        for _ in range(max_new_tokens):
            nxt = sample_next(logits, temperature, top_k, top_p, gen)
            out.append(nxt.unsqueeze(1))
            logits = self._forward(nxt.unsqueeze(1), kc, vc, pos)
            pos += 1
        return torch.cat(out, dim=1)
