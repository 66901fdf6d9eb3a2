"""factorvae_amd — MI355X-native FactorVAE training/inference engine.

A from-scratch re-design of the capabilities of x7jeon8gi/FactorVAE
(Duan et al., AAAI 2022 — probabilistic dynamic factor model for
cross-sectional stock returns) for AMD Instinct MI355X (gfx950, CDNA4):

- PyTorch-ROCm front-end with the same public module API and checkpoint
  contract as the reference (`/root/reference/module.py:10-278`,
  `main.py:78-79`).
- Hand-written HIP/CDNA4 kernels for every hot op (fused
  LayerNorm+Linear+LeakyReLU+GRU extractor, stock-axis softmax encoder,
  algebraically-restructured K-head cross-sectional attention, fused
  alpha/beta decoder + reparameterization, fused MSE+KL loss, fused Adam).
- Data parallelism over trading-day mini-batches with RCCL all-reduce
  over xGMI (one flat gradient bucket).
- Device-resident epoch cache sized for 288 GB HBM3E per GPU.
"""

__version__ = "0.2.0"

from .models.modules import (
    FeatureExtractor,
    FactorEncoder,
    AlphaLayer,
    BetaLayer,
    FactorDecoder,
    AttentionLayer,
    FactorPredictor,
    FactorVAE,
)
from .utils import (
    set_seed,
    DataArgument,
    test_args,
    load_model,
    generate_prediction_scores,
    RankIC,
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
    "set_seed",
    "DataArgument",
    "test_args",
    "load_model",
    "generate_prediction_scores",
    "RankIC",
]
