"""hetu_amd.utils — checkpointing, HF interop, parallel-config files,
profiling, logging (reference python/hetu/utils)."""
from .checkpoint import load_model, save_model  # noqa: F401
from .logging import get_logger  # noqa: F401
from .profiler import MemorySnapshots, OpProfiler  # noqa: F401
