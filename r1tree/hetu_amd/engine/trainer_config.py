"""Structured training config + YAML loading (reference
engine/trainer_config.py + the hydra-style examples/pretrain/config/*.yaml
files: model architecture name, parallel strategy, batching, precision)."""
from __future__ import annotations

import dataclasses
from typing import Optional

import yaml


@dataclasses.dataclass
class TrainingConfig:
    architecture: str = "GPTLMHeadModel"      # or LlamaLMHeadModel
    model: str = "gpt3-7b"
    seq_len: int = 2048
    global_batch: int = 8
    micro_batch: int = 1
    dp: int = 1
    tp: int = 1
    pp: int = 1
    cp: int = 1
    zero: bool = False
    sequence_parallel: bool = False
    recompute: bool = False
    precision: str = "bf16"                   # bf16 | fp32 | fp16(+scaler)
    lr: float = 1e-4
    weight_decay: float = 0.0
    steps: int = 100
    save_every: int = 0
    save_path: Optional[str] = None
    seed: int = 1234

    @classmethod
    def from_yaml(cls, path: str) -> "TrainingConfig":
        with open(path) as f:
            raw = yaml.safe_load(f) or {}
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in raw.items() if k in known})

    def dtype(self):
        import torch
        return {"bf16": torch.bfloat16, "fp16": torch.float16,
                "fp32": torch.float32}[self.precision]
