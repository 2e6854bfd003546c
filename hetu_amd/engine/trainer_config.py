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
    lr_warmup_steps: int = 0
    lr_decay: str = "none"            # none | cosine | inv_sqrt
    lr_min_ratio: float = 0.1
    weight_decay: float = 0.0
    steps: int = 100
    save_every: int = 0
    save_path: Optional[str] = None
    seed: int = 1234
    # path to a layered JSON ds_parallel_config (reference generate_ds.py
    # format): when set, dp/tp/pp/zero are read from the file instead
    ds_parallel_config: Optional[str] = None

    @classmethod
    def from_yaml(cls, path: str) -> "TrainingConfig":
        with open(path) as f:
            raw = yaml.safe_load(f) or {}
        known = {f.name for f in dataclasses.fields(cls)}
        tc = cls(**{k: v for k, v in raw.items() if k in known})
        if tc.ds_parallel_config:
            tc.apply_ds_config(tc.ds_parallel_config)
        return tc

    def apply_ds_config(self, path: str) -> "TrainingConfig":
        """Override the parallel strategy from a ds_parallel_config JSON
        (homogeneous configs only; hetero worlds use HeteroSpec directly)."""
        import json

        from ..parallel.pipeline import PipelineSpec
        from ..utils.ds_config import strategy_from_config
        spec, _ = strategy_from_config(path)
        if not isinstance(spec, PipelineSpec):
            raise ValueError(
                "ds_parallel_config is heterogeneous: drive it through "
                "parallel.hetero.HeteroSpec (examples/malleus)")
        self.pp, self.dp, self.tp = spec.pp, spec.dp, spec.tp
        with open(path) as f:
            self.zero = bool(json.load(f).get("zero", False))
        return self

    def lr_schedule(self):
        """Multiplier schedule for the Trainer (None when static)."""
        from .lr_schedule import (cosine_with_warmup, inverse_sqrt,
                                  linear_warmup)
        if self.lr_decay == "cosine":
            return cosine_with_warmup(self.lr_warmup_steps, self.steps,
                                      self.lr_min_ratio)
        if self.lr_decay == "inv_sqrt":
            return inverse_sqrt(max(1, self.lr_warmup_steps))
        if self.lr_warmup_steps > 0:
            return linear_warmup(self.lr_warmup_steps)
        return None

    def dtype(self):
        import torch
        return {"bf16": torch.bfloat16, "fp16": torch.float16,
                "fp32": torch.float32}[self.precision]
