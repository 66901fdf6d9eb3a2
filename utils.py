"""Drop-in root module matching the reference layout
(/root/reference/utils.py). Implementations: factorvae_amd/utils.py.
"""
from factorvae_amd.utils import (
    DataArgument,
    RankIC,
    generate_prediction_scores,
    load_model,
    set_seed,
    test_args,
)

__all__ = ["set_seed", "DataArgument", "load_model",
           "generate_prediction_scores", "test_args", "RankIC"]
