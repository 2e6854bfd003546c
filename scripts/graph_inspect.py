#!/usr/bin/env python3
"""Inspect the captured hipGraph's node/dependency structure via ctypes:
single-stream capture must yield a linear chain (every node except the
first has >=1 dependency).  Zero-dependency nodes would launch concurrently
at replay — corruption."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

hip = ctypes.CDLL("libamdhip64.so")
hip.hipGraphGetNodes.argtypes = [ctypes.c_void_p,
                                 ctypes.POINTER(ctypes.c_void_p),
                                 ctypes.POINTER(ctypes.c_size_t)]
hip.hipGraphNodeGetType.argtypes = [ctypes.c_void_p,
                                    ctypes.POINTER(ctypes.c_int)]
hip.hipGraphNodeGetDependencies.argtypes = [ctypes.c_void_p,
                                            ctypes.POINTER(ctypes.c_void_p),
                                            ctypes.POINTER(ctypes.c_size_t)]

NODE_TYPES = {0: "Kernel", 1: "Memcpy", 2: "Memset", 3: "Host",
              4: "Graph", 5: "Empty", 6: "WaitEvent", 7: "EventRecord",
              8: "ExtSemSignal", 9: "ExtSemWait", 10: "MemAlloc",
              11: "MemFree", 12: "MemcpyFromSymbol", 13: "MemcpyToSymbol"}


def inspect(cg, tag):
    h = ctypes.c_void_p(cg.raw_cuda_graph())
    n = ctypes.c_size_t(0)
    hip.hipGraphGetNodes(h, None, ctypes.byref(n))
    cnt = n.value
    nodes = (ctypes.c_void_p * cnt)()
    hip.hipGraphGetNodes(h, nodes, ctypes.byref(n))
    types = {}
    zero_dep = 0
    dep_hist = {}
    for i in range(cnt):
        t = ctypes.c_int(-1)
        hip.hipGraphNodeGetType(ctypes.c_void_p(nodes[i]), ctypes.byref(t))
        types[NODE_TYPES.get(t.value, t.value)] = \
            types.get(NODE_TYPES.get(t.value, t.value), 0) + 1
        nd = ctypes.c_size_t(0)
        hip.hipGraphNodeGetDependencies(ctypes.c_void_p(nodes[i]), None,
                                        ctypes.byref(nd))
        dep_hist[nd.value] = dep_hist.get(nd.value, 0) + 1
        if nd.value == 0:
            zero_dep += 1
    print(f"[{tag}] nodes={cnt} types={types} zero_dep={zero_dep} "
          f"dep_hist={dict(sorted(dep_hist.items()))}", flush=True)


def main():
    from hetu_amd.engine.trainer import Trainer
    from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph
    dev = torch.device("cuda", 0)
    L, H, B, S, V = 12, 256, 2, 256, 50304
    torch.manual_seed(1234)
    cfg = GPTConfig(n_layer=L, n_head=2, n_kv_head=2, hidden=H,
                    ffn_hidden=4 * H, vocab=V, max_seq=S)
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16, lr=1e-4)
    tr = Trainer(g, h, dev, capture=False)
    feed = {h["input_ids"]: torch.randint(0, V, (B, S), device=dev),
            h["labels"]: torch.randint(0, V, (B * S,), device=dev)}
    tr.run_step(dict(feed))
    torch.cuda.synchronize()
    for t, v in feed.items():
        tr._static_feeds[t] = v.to(dev).clone()
    cg = torch.cuda.CUDAGraph(keep_graph=True)
    kept = {} if os.environ.get("KEEPVALS") == "1" else None
    with torch.cuda.graph(cg):
        tr._loss_out = g.run([h["loss"], h["train_op"]],
                             dict(tr._static_feeds), ctx=tr.ctx,
                             keep_values=kept)[0]
    inspect(cg, "real-gpt-L12-h256")
    # and replay a few to confirm this capture also corrupts
    tr._cuda_graph = cg
    adams = [op for op in g.ops if op.type == "AdamStep"]
    for i in range(4):
        tr.replay()
        torch.cuda.synchronize()
        nb = sum(1 for op in adams
                 if not torch.isfinite(op.interface.state["m"]).all())
        print(f"replay {i}: loss={float(tr._loss_out.float()):.4f} "
              f"bad={nb}", flush=True)


if __name__ == "__main__":
    main()
