"""hetu_amd.optim — optimizer API namespace (reference python/hetu/optim).

The update ops themselves live in graph.ops.optim (they are graph ops);
this package mirrors the reference's import surface and adds the LR
schedules."""
from ..engine.lr_schedule import (constant, cosine_with_warmup,  # noqa
                                  inverse_sqrt, linear_warmup)
from ..graph.ops.optim import SGD, Adam, Optimizer  # noqa: F401
