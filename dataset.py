"""Drop-in root module matching the reference layout
(/root/reference/dataset.py). Implementations:
factorvae_amd/data/sampler.py.
"""
from factorvae_amd.data.sampler import (
    DateGroupedBatchSampler,
    TSDataSampler,
    TSDatasetH,
    custom_collate_fn,
    init_data_loader,
    np_ffill,
)

__all__ = ["np_ffill", "TSDataSampler", "TSDatasetH",
           "DateGroupedBatchSampler", "custom_collate_fn",
           "init_data_loader"]
