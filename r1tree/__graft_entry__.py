"""Driver contract: build() compiles the HIP extensions for gfx950;
smoke() runs one tiny forward+backward of the flagship model on cuda:0."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def build() -> None:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from hetu_amd.ops.hip.build import build as _build
    _build()
    import hetu_amd  # noqa: F401
    import hetu_amd.ops.functional as F
    assert F.has_ext(), "HIP extension failed to load after build"


def smoke() -> None:
    import torch
    import hetu_amd as ht
    from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph

    assert torch.cuda.is_available(), "smoke() requires a GPU"
    cfg = GPTConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=512,
                    ffn_hidden=2048, vocab=50304, max_seq=256)
    g, handles = build_gpt_train_graph(cfg, micro_batch=2, seq_len=256,
                                       dtype=torch.bfloat16)
    from hetu_amd.engine.runner import prepare_run_context
    ctx = prepare_run_context(g, torch.device("cuda", 0))
    ids = torch.randint(0, cfg.vocab, (2, 256), device="cuda")
    labels = torch.randint(0, cfg.vocab, (2 * 256,), device="cuda")
    loss, _ = g.run([handles["loss"], handles["train_op"]],
                    {handles["input_ids"]: ids, handles["labels"]: labels},
                    ctx=ctx)
    torch.cuda.synchronize()
    lv = float(loss.item())
    assert lv == lv and lv > 0, f"bad loss {lv}"
    print(f"smoke OK: loss={lv:.4f}")


if __name__ == "__main__":
    build()
    if len(sys.argv) > 1 and sys.argv[1] == "smoke":
        smoke()
