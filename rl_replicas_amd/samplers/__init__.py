from .sampler import Sampler
from .batch_sampler import BatchSampler
from .device_sampler import DeviceSampler
from .vector_sampler import VectorSampler

__all__ = ["Sampler", "BatchSampler", "VectorSampler", "DeviceSampler"]
