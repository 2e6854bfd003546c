"""nn.Module system for the define-and-run graph.

Reference parity: python/hetu/nn/modules/module.py (Module, parameter
registry, state_dict/_load_from_state_dict) and container.py (Sequential,
ModuleList).  MI355X-native design difference: parameters are graph
`variable` tensors created inside the currently-active graph at module
construction time; their payloads are torch tensors (the local shard when a
DistributedStates annotation is present), so state_dict round-trips through
safetensors without any custom tensor type.
"""
from __future__ import annotations

from collections import OrderedDict
from typing import Dict, Iterator, List, Optional, Tuple

import torch

from ..graph.tensor import Tensor


class Module:
    def __init__(self):
        object.__setattr__(self, "_parameters", OrderedDict())
        object.__setattr__(self, "_modules", OrderedDict())
        object.__setattr__(self, "_buffers", OrderedDict())
        self.training = True

    # ---- registration ----------------------------------------------------
    def __setattr__(self, name, value):
        if isinstance(value, Tensor) and getattr(value, "is_parameter", False):
            self._parameters[name] = value
        elif isinstance(value, Module):
            self._modules[name] = value
        object.__setattr__(self, name, value)

    def register_parameter(self, name: str, param: Optional[Tensor]):
        self._parameters[name] = param
        object.__setattr__(self, name, param)

    def register_buffer(self, name: str, t):
        self._buffers[name] = t
        object.__setattr__(self, name, t)

    def add_module(self, name: str, module: "Module"):
        self._modules[name] = module
        object.__setattr__(self, name, module)

    # ---- traversal ---------------------------------------------------------
    def named_parameters(self, prefix: str = ""
                         ) -> Iterator[Tuple[str, Tensor]]:
        for n, p in self._parameters.items():
            if p is not None:
                yield (prefix + n if not prefix else f"{prefix}.{n}"), p
        for n, m in self._modules.items():
            sub = n if not prefix else f"{prefix}.{n}"
            yield from m.named_parameters(sub)

    def parameters(self) -> List[Tensor]:
        return [p for _, p in self.named_parameters()]

    def named_modules(self, prefix: str = ""):
        yield prefix, self
        for n, m in self._modules.items():
            sub = n if not prefix else f"{prefix}.{n}"
            yield from m.named_modules(sub)

    def modules(self):
        return (m for _, m in self.named_modules())

    def children(self):
        return iter(self._modules.values())

    # ---- mode --------------------------------------------------------------
    def train(self, mode: bool = True):
        self.training = mode
        for m in self._modules.values():
            m.train(mode)
        return self

    def eval(self):
        return self.train(False)

    # ---- state dict --------------------------------------------------------
    def state_dict(self, prefix: str = "") -> "OrderedDict[str, torch.Tensor]":
        """Local-shard state dict (torch tensors).  DS-aware global
        assembly lives in utils/checkpoint (ht_safetensors parity)."""
        out: "OrderedDict[str, torch.Tensor]" = OrderedDict()
        for name, p in self.named_parameters(prefix):
            out[name] = p.get_data()
        return out

    def load_state_dict(self, sd: Dict[str, torch.Tensor], strict: bool = True):
        missing, unexpected = [], []
        mine = dict(self.named_parameters())
        for name, p in mine.items():
            if name in sd:
                data = sd[name]
                cur = p.get_data()
                if cur is not None and tuple(cur.shape) != tuple(data.shape):
                    raise ValueError(
                        f"shape mismatch for {name}: "
                        f"{tuple(cur.shape)} vs {tuple(data.shape)}")
                if cur is not None:
                    cur.copy_(data.to(cur.dtype).to(cur.device))
                else:
                    p.set_data(data)
            else:
                missing.append(name)
        for name in sd:
            if name not in mine:
                unexpected.append(name)
        if strict and (missing or unexpected):
            raise KeyError(f"missing={missing} unexpected={unexpected}")
        return missing, unexpected

    # ---- call --------------------------------------------------------------
    def forward(self, *args, **kwargs):
        raise NotImplementedError

    def __call__(self, *args, **kwargs):
        return self.forward(*args, **kwargs)

    def __repr__(self):
        lines = [self.__class__.__name__ + "("]
        for n, m in self._modules.items():
            sub = repr(m).replace("\n", "\n  ")
            lines.append(f"  ({n}): {sub}")
        lines.append(")")
        return "\n".join(lines)


class Sequential(Module):
    def __init__(self, *mods):
        super().__init__()
        if len(mods) == 1 and isinstance(mods[0], OrderedDict):
            for n, m in mods[0].items():
                self.add_module(n, m)
        else:
            for i, m in enumerate(mods):
                self.add_module(str(i), m)

    def __iter__(self):
        return iter(self._modules.values())

    def __len__(self):
        return len(self._modules)

    def __getitem__(self, idx):
        return list(self._modules.values())[idx]

    def forward(self, x):
        for m in self._modules.values():
            x = m(x)
        return x


class ModuleList(Module):
    def __init__(self, mods=()):
        super().__init__()
        for i, m in enumerate(mods):
            self.add_module(str(i), m)

    def append(self, m: Module):
        self.add_module(str(len(self._modules)), m)
        return self

    def __iter__(self):
        return iter(self._modules.values())

    def __len__(self):
        return len(self._modules)

    def __getitem__(self, idx):
        return list(self._modules.values())[idx]


class ModuleDict(Module):
    def __init__(self, mods: Optional[Dict[str, Module]] = None):
        super().__init__()
        for n, m in (mods or {}).items():
            self.add_module(n, m)

    def __getitem__(self, key):
        return self._modules[key]

    def __contains__(self, key):
        return key in self._modules

    def keys(self):
        return self._modules.keys()

    def items(self):
        return self._modules.items()
