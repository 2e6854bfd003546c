"""Supervised fine-tuning trainer (reference engine/sft_trainer.py):
prompt/completion pairs with the prompt tokens masked out of the loss
(ignore_index), running on the same define-and-run train graph."""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

import torch

from .trainer import Trainer


def build_sft_example(prompt_ids: List[int], answer_ids: List[int],
                      seq_len: int, pad_token: int = 0,
                      ignore_index: int = -100
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """(input_ids [S], labels [S]) with the prompt and padding masked."""
    ids = (prompt_ids + answer_ids)[:seq_len + 1]
    x = torch.full((seq_len,), pad_token, dtype=torch.int64)
    y = torch.full((seq_len,), ignore_index, dtype=torch.int64)
    inp = ids[:-1][:seq_len]
    tgt = ids[1:][:seq_len]
    x[:len(inp)] = torch.tensor(inp)
    y[:len(tgt)] = torch.tensor(tgt)
    # mask prompt positions (targets that belong to the prompt)
    n_prompt = max(0, min(len(prompt_ids) - 1, seq_len))
    y[:n_prompt] = ignore_index
    return x, y


class SFTTrainer(Trainer):
    """Trainer whose step() takes (input_ids [B,S], labels [B,S]) built by
    build_sft_example; the vocab-parallel CE already honors ignore_index."""

    def sft_step(self, input_ids: torch.Tensor, labels: torch.Tensor):
        return self.step({self.h["input_ids"]: input_ids,
                          self.h["labels"]: labels.reshape(-1)})
