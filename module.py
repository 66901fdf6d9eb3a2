"""Drop-in root module matching the reference layout
(/root/reference/module.py): `from module import FactorVAE, ...` works
unchanged for reference users. Implementations:
factorvae_amd/models/modules.py. (`FactorVAE_old` is the reference's
dead/broken variant — intentionally not carried, SURVEY.md §2.1 #16.)
"""
from factorvae_amd.models.modules import (
    AlphaLayer,
    AttentionLayer,
    BetaLayer,
    FactorDecoder,
    FactorEncoder,
    FactorPredictor,
    FactorVAE,
    FeatureExtractor,
)

__all__ = ["FeatureExtractor", "FactorEncoder", "AlphaLayer", "BetaLayer",
           "FactorDecoder", "AttentionLayer", "FactorPredictor", "FactorVAE"]
