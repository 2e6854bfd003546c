"""hetu_amd.data — datasets, packing, samplers, loaders
(reference python/hetu/data)."""
from .bucket import Bucket  # noqa: F401
from .dataloader import (JsonDataset, PrefetchLoader,  # noqa: F401
                         SampleBatchSampler, SyntheticLMDataset,
                         TokenBatchSampler)
