from .sampler import (
    np_ffill,
    TSDataSampler,
    TSDatasetH,
    DateGroupedBatchSampler,
    custom_collate_fn,
    init_data_loader,
)
from .synthetic import make_synthetic_frame, N_ALPHA_FEATURES
from .device_cache import DeviceEpochCache

__all__ = [
    "np_ffill",
    "TSDataSampler",
    "TSDatasetH",
    "DateGroupedBatchSampler",
    "custom_collate_fn",
    "init_data_loader",
    "make_synthetic_frame",
    "N_ALPHA_FEATURES",
    "DeviceEpochCache",
]
