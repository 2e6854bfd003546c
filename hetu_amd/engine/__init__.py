"""hetu_amd.engine — trainers, schedules, elastic/dynamic planners
(reference python/hetu/engine)."""
from .lr_schedule import (constant, cosine_with_warmup,  # noqa: F401
                          inverse_sqrt, linear_warmup)
from .trainer import Trainer  # noqa: F401
from .trainer_config import TrainingConfig  # noqa: F401
