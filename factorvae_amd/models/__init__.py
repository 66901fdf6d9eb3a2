from .modules import (
    FeatureExtractor,
    FactorEncoder,
    AlphaLayer,
    BetaLayer,
    FactorDecoder,
    AttentionLayer,
    FactorPredictor,
    FactorVAE,
    build_factorvae,
)

__all__ = [
    "FeatureExtractor",
    "FactorEncoder",
    "AlphaLayer",
    "BetaLayer",
    "FactorDecoder",
    "AttentionLayer",
    "FactorPredictor",
    "FactorVAE",
    "build_factorvae",
]
