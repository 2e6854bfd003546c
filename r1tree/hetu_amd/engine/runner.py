"""Run-context helpers: bind a graph's variables + executor to a device and
the RCCL comm backend."""
from __future__ import annotations

import torch

from ..graph.executor import ExecContext
from ..parallel.comm import comm_backend


def prepare_run_context(graph, device: torch.device,
                        use_comm: bool = True) -> ExecContext:
    """Move all variable storage to `device`, init comm (if distributed),
    and bind an ExecContext to the graph's executor."""
    comm = comm_backend(device) if use_comm else None
    for t in list(graph._tensor_by_id.values()):
        data = t.get_data()
        if data is not None and data.device != device:
            t.set_data(data.to(device))
    ctx = ExecContext(device=device, comm=comm)
    if hasattr(graph, "executor"):
        graph.executor().bind_context(ctx)
    return ctx
