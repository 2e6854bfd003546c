"""Loss-curve parity vs a pure-PyTorch model (reference
examples/hetero/train_pytorch_gpt.py methodology): the same GPT, same
weights, same data, trained by our graph engine and by plain torch autograd
+ torch.optim.Adam must produce the same loss curve in fp32."""
import math

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as TF


class TorchGPT(nn.Module):
    """Mirror of hetu_amd.models.gpt (pre-LN, gelu-tanh MLP, learned
    positions, causal attention)."""

    def __init__(self, cfg, seq_len):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab, cfg.hidden)
        self.wpe = nn.Embedding(seq_len, cfg.hidden)
        self.blocks = nn.ModuleList()
        for _ in range(cfg.n_layer):
            blk = nn.ModuleDict({
                "ln1": nn.LayerNorm(cfg.hidden, eps=1e-5),
                "wqkv": nn.Linear(cfg.hidden, 3 * cfg.hidden),
                "wo": nn.Linear(cfg.hidden, cfg.hidden),
                "ln2": nn.LayerNorm(cfg.hidden, eps=1e-5),
                "wfc": nn.Linear(cfg.hidden, cfg.ffn_hidden),
                "wproj": nn.Linear(cfg.ffn_hidden, cfg.hidden),
            })
            self.blocks.append(blk)
        self.lnf = nn.LayerNorm(cfg.hidden, eps=1e-5)
        self.lm_head = nn.Linear(cfg.hidden, cfg.vocab, bias=False)

    def forward(self, ids, labels):
        B, S = ids.shape
        H = self.cfg.n_head
        dh = self.cfg.hidden // H
        x = self.wte(ids) + self.wpe(torch.arange(S))
        for blk in self.blocks:
            h = blk["ln1"](x)
            qkv = blk["wqkv"](h)
            q, k, v = qkv.chunk(3, -1)
            q = q.view(B, S, H, dh).transpose(1, 2)
            k = k.view(B, S, H, dh).transpose(1, 2)
            v = v.view(B, S, H, dh).transpose(1, 2)
            o = TF.scaled_dot_product_attention(q, k, v, is_causal=True)
            o = o.transpose(1, 2).reshape(B, S, -1)
            x = x + blk["wo"](o)
            h = blk["ln2"](x)
            h = TF.gelu(blk["wfc"](h), approximate="tanh")
            x = x + blk["wproj"](h)
        logits = self.lm_head(self.lnf(x))
        return TF.cross_entropy(logits.reshape(-1, self.cfg.vocab),
                                labels)


def test_gpt_loss_curve_matches_pytorch():
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph
    torch.manual_seed(0)
    cfg = GPTConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=64,
                    ffn_hidden=128, vocab=211, max_seq=32)
    B, S, lr = 2, 32, 1e-3
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.float32, lr=lr)
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)

    ref = TorchGPT(cfg, S)
    # copy OUR initialization into the torch model
    params = {p.name.split(":")[0]: p.get_data() for p in g.parameters}
    sd = {}
    sd["wte.weight"] = params["wte.weight"]
    sd["wpe.weight"] = params["wpe.weight"]
    sd["lnf.weight"] = params["lnf.weight"]
    sd["lnf.bias"] = params["lnf.bias"]
    sd["lm_head.weight"] = params["lm_head.weight"]
    for i in range(cfg.n_layer):
        for src, dst in [
            (f"h{i}.ln1", f"blocks.{i}.ln1"),
            (f"h{i}.ln2", f"blocks.{i}.ln2"),
            (f"h{i}.attn.wqkv", f"blocks.{i}.wqkv"),
            (f"h{i}.attn.wo", f"blocks.{i}.wo"),
            (f"h{i}.mlp.wfc", f"blocks.{i}.wfc"),
            (f"h{i}.mlp.wproj", f"blocks.{i}.wproj"),
        ]:
            sd[f"{dst}.weight"] = params[f"{src}.weight"]
            if f"{src}.bias" in params:
                sd[f"{dst}.bias"] = params[f"{src}.bias"]
    missing, unexpected = ref.load_state_dict(sd, strict=True), None
    opt = torch.optim.Adam(ref.parameters(), lr=lr, betas=(0.9, 0.999),
                           eps=1e-8)

    gen = torch.Generator().manual_seed(7)
    ours, theirs = [], []
    for step in range(5):
        ids = torch.randint(0, cfg.vocab, (B, S), generator=gen)
        labels = torch.randint(0, cfg.vocab, (B * S,), generator=gen)
        lv, _ = g.run([h["loss"], h["train_op"]],
                      {h["input_ids"]: ids, h["labels"]: labels}, ctx=ctx)
        ours.append(float(lv))
        loss = ref(ids, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        theirs.append(float(loss.detach()))
    for a, b in zip(ours, theirs):
        assert math.isfinite(a) and abs(a - b) < 5e-3, (ours, theirs)


def test_lr_schedule_matches_torch_lambda_lr():
    """Our multiplier schedules must track torch.optim LambdaLR exactly:
    same tiny model, same cosine-with-warmup, identical loss curves."""
    from hetu_amd.engine.lr_schedule import cosine_with_warmup
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.ops.optim import AdamStepOp
    from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.graph.ops.optim import Adam

    torch.manual_seed(4)
    d = 16
    w0 = torch.randn(d, d) * 0.3
    sched = cosine_with_warmup(2, 8, min_ratio=0.1)
    lr = 5e-3

    g = DefineAndRunGraph("lrp")
    push_graph(g)
    try:
        x = ht.placeholder((4, d), name="x")
        t = ht.placeholder((4, d), name="t")
        w = ht.variable(w0.clone(), name="w")
        loss = ht.mse_loss(ht.matmul(x, w), t)
        train = Adam(lr=lr).minimize(loss)
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)

    wr = w0.clone().requires_grad_(True)
    opt = torch.optim.Adam([wr], lr=lr, betas=(0.9, 0.999), eps=1e-8)
    lsched = torch.optim.lr_scheduler.LambdaLR(opt, sched)

    gen = torch.Generator().manual_seed(9)
    try:
        for step in range(8):
            xd = torch.randn(4, d, generator=gen)
            td = torch.randn(4, d, generator=gen)
            AdamStepOp.set_lr_scale(sched(step))
            lv, _ = g.run([loss, train], {x: xd, t: td}, ctx=ctx)
            rl = torch.nn.functional.mse_loss(xd @ wr, td)
            opt.zero_grad(); rl.backward(); opt.step(); lsched.step()
            assert abs(float(lv) - float(rl)) < 1e-5, (step, float(lv),
                                                       float(rl))
        assert torch.allclose(w.get_data(), wr.detach(), atol=1e-5)
    finally:
        AdamStepOp.set_lr_scale(1.0)
