from .sampler import Sampler
from .batch_sampler import BatchSampler
from .vector_sampler import VectorSampler

__all__ = ["Sampler", "BatchSampler", "VectorSampler"]
