"""hetu_amd.nn — module system + layers (reference python/hetu/nn)."""
from . import init  # noqa: F401
from .module import Module, ModuleList, Sequential  # noqa: F401
from .modules import (BCELoss, AvgPool2d, BatchNorm2d, Conv2d,  # noqa: F401
                      CrossEntropyLoss, Dropout, Embedding, GELU,
                      InstanceNorm2d, KLDivLoss, LayerNorm, Linear,
                      MaxPool2d, MSELoss, NLLLoss, ReLU, RMSNorm,
                      Sigmoid, SiLU, Tanh, ZeroPad2d)
from .parallel import (ColumnParallelLinear, ParallelSpec,  # noqa: F401
                       ParallelLayerNorm, ParallelRMSNorm,
                       RowParallelLinear, VocabParallelEmbedding)
