import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.elastic_loop import run_elastic_training  # noqa
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph  # noqa
from hetu_amd.nn.parallel import ParallelSpec  # noqa: E402

rank = int(os.environ["ELASTIC_RANK"])
world = int(os.environ["ELASTIC_WORLD"])
cfg = GPTConfig(n_layer=2, n_head=2, n_kv_head=2, hidden=64,
                ffn_hidden=128, vocab=128, max_seq=16)


def build_fn(ws, comm):
    spec = ParallelSpec(dp=ws, device_group=list(range(ws)))
    return build_gpt_train_graph(cfg, micro_batch=2, seq_len=16,
                                 dtype=torch.float32, lr=1e-3, spec=spec)


def feed_fn(step):
    gen = torch.Generator().manual_seed(1000 + 31 * step + rank)
    return (torch.randint(0, 128, (2, 16), generator=gen),
            torch.randint(0, 128, (32,), generator=gen))


die_at = int(os.environ.get("DIE_AT", "-1"))
res = run_elastic_training(
    build_fn, feed_fn, total_steps=int(os.environ.get("STEPS", "8")),
    ckpt_dir=os.environ.get("CKPT_DIR", "/tmp/elastic_ckpt"),
    kv_host="127.0.0.1", kv_port=int(os.environ["KV_PORT"]),
    rank=rank, world=world, heartbeat_timeout_s=2.0,
    die_at=die_at if die_at >= 0 else None)
print(f"rank {rank}: {json.dumps(res)}")
