"""hetu_amd.parallel — DS algebra, comm backend, pipelines, rings,
hot switch, hetero DP (reference hetu/graph distributed layer).

Only the leaf modules are re-exported here: graph.tensor imports
parallel.dstates at package-import time, so pulling pipeline/hetero (which
import nn/graph) into this __init__ would create an import cycle — import
those submodules directly (`from hetu_amd.parallel.pipeline import ...`).
"""
from .comm import CommBackend, comm_backend  # noqa: F401
from .dstates import (DistributedStates,  # noqa: F401
                      DistributedStatesUnion, NULL_HETERO_DIM)
