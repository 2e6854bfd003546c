"""hetu_amd — MI355X-native distributed deep-learning framework.

A from-scratch rebuild of PKU-DAIR/Hetu's capabilities for AMD Instinct
MI355X (gfx950): define-and-run SPMD-annotated dataflow graphs executing on
PyTorch-ROCm tensors, hand-written CDNA4 HIP kernels for the hot ops, and
RCCL-over-xGMI collectives via torch.distributed.
"""
from . import ops as _kernel_ops  # noqa: F401
from .core.symbol import IntSymbol
from .graph.graph import (DefineAndRunGraph, EagerGraph, Graph,
                          current_graph, graph)
from .graph.tensor import Tensor
from .graph.ops.api import *  # noqa: F401,F403
from .graph.ops.api import comm, gradients, placeholder, variable
from .graph.ops.optim import SGD, Adam, Optimizer
from .parallel.dstates import (DistributedStates, ds_dup, ds_partial,
                               ds_split)
from .parallel.comm import CommBackend, comm_backend
from .utils.profiler import MemorySnapshots, OpProfiler

__version__ = "0.1.0"
