"""FactorVAE model modules — public API compatible with the reference.

Class surface, tensor semantics and checkpoint key namespace match
`/root/reference/module.py:10-278` exactly (verified by parity tests);
the implementation is new and MI355X-first:

- the eager path below is the *oracle*: plain PyTorch ops reproducing the
  reference math bit-for-bit (softmax over the stock axis dim=0,
  dropout-before-ReLU-before-softmax in attention, NaN-guard zero
  contexts, in-place sigma clamps, sum-reduced KL, stochastic decoder at
  inference);
- the fused GPU path (factorvae_amd.engine) re-expresses the same math as
  a handful of hand-written HIP/CDNA4 kernels and is validated against
  this oracle.

Shape glossary: N stocks/day, T lookback, C=158 features,
H hidden, M portfolios, K factors.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class FeatureExtractor(nn.Module):
    """Per-stock temporal encoder: LayerNorm(C) -> Linear(C,C) -> LeakyReLU
    -> GRU(C,H) -> last timestep.

    Reference: /root/reference/module.py:10-31. Checkpoint keys:
    `normalize.{weight,bias}`, `linear.{weight,bias}`,
    `gru.{weight_ih_l0,weight_hh_l0,bias_ih_l0,bias_hh_l0}` (gate order
    r,z,n preserved).
    """

    def __init__(self, num_latent: int, hidden_size: int, num_layers: int = 1):
        super().__init__()
        self.num_latent = num_latent
        self.hidden_size = hidden_size
        self.num_layers = num_layers

        self.normalize = nn.LayerNorm(num_latent)
        self.linear = nn.Linear(num_latent, num_latent)
        self.leakyrelu = nn.LeakyReLU()
        self.gru = nn.GRU(num_latent, hidden_size, num_layers, batch_first=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # x: (N, T, C) -> (N, H)
        x = self.normalize(x)
        out = self.leakyrelu(self.linear(x))
        stock_latent, _ = self.gru(out)
        return stock_latent[:, -1, :]


class FactorEncoder(nn.Module):
    """Posterior q(z|x,y): portfolio weights softmax over the STOCK axis
    (dim=0), portfolio returns Wᵀy, then mu/softplus-sigma heads.

    Reference: /root/reference/module.py:33-67.
    """

    def __init__(self, num_factors: int, num_portfolio: int, hidden_size: int):
        super().__init__()
        self.num_factors = num_factors
        self.linear = nn.Linear(hidden_size, num_portfolio)
        self.softmax = nn.Softmax(dim=0)

        self.linear_mu = nn.Linear(num_portfolio, num_factors)
        self.linear_sigma = nn.Linear(num_portfolio, num_factors)
        self.softplus = nn.Softplus()

    def mapping_layer(self, portfolio_return: torch.Tensor):
        # portfolio_return: (M, 1) -> (K,), (K,)
        mean = self.linear_mu(portfolio_return.squeeze(1))
        sigma = self.softplus(self.linear_sigma(portfolio_return.squeeze(1)))
        return mean, sigma

    def forward(self, stock_latent: torch.Tensor, returns: torch.Tensor):
        # stock_latent: (N, H); returns: (N, 1) -> mu (K,), sigma (K,)
        weights = self.softmax(self.linear(stock_latent))  # (N, M), softmax over stocks
        if returns.dim() == 1:
            returns = returns.unsqueeze(1)
        portfolio_return = torch.mm(weights.transpose(1, 0), returns)  # (M, 1)
        return self.mapping_layer(portfolio_return)


class AlphaLayer(nn.Module):
    """Idiosyncratic return head. Reference: /root/reference/module.py:69-84."""

    def __init__(self, hidden_size: int):
        super().__init__()
        self.linear1 = nn.Linear(hidden_size, hidden_size)
        self.leakyrelu = nn.LeakyReLU()
        self.mu_layer = nn.Linear(hidden_size, 1)
        self.sigma_layer = nn.Linear(hidden_size, 1)
        self.softplus = nn.Softplus()

    def forward(self, stock_latent: torch.Tensor):
        h = self.leakyrelu(self.linear1(stock_latent))
        alpha_mu = self.mu_layer(h)
        alpha_sigma = self.softplus(self.sigma_layer(h))
        return alpha_mu, alpha_sigma


class BetaLayer(nn.Module):
    """Factor exposures beta (N,K). Reference: /root/reference/module.py:86-94."""

    def __init__(self, hidden_size: int, num_factors: int):
        super().__init__()
        self.linear1 = nn.Linear(hidden_size, num_factors)

    def forward(self, stock_latent: torch.Tensor) -> torch.Tensor:
        return self.linear1(stock_latent)


class FactorDecoder(nn.Module):
    """Return decoder: mu = alpha_mu + beta @ z_mu;
    sigma = sqrt(alpha_sigma^2 + beta^2 @ z_sigma^2 + 1e-6);
    output is SAMPLED: y = mu + eps * sigma (stochastic even at eval).

    Reference: /root/reference/module.py:96-123 (incl. the in-place
    factor_sigma==0 -> 1e-6 clamp at :117).
    """

    def __init__(self, alpha_layer: AlphaLayer, beta_layer: BetaLayer):
        super().__init__()
        self.alpha_layer = alpha_layer
        self.beta_layer = beta_layer

    def reparameterize(self, mu: torch.Tensor, sigma: torch.Tensor) -> torch.Tensor:
        eps = torch.randn_like(sigma)
        return mu + eps * sigma

    def forward(self, stock_latent, factor_mu, factor_sigma):
        alpha_mu, alpha_sigma = self.alpha_layer(stock_latent)
        beta = self.beta_layer(stock_latent)

        factor_mu = factor_mu.view(-1, 1)
        factor_sigma = factor_sigma.view(-1, 1)
        # clamp exact zeros to 1e-6 (reference does an in-place masked store)
        factor_sigma = torch.where(
            factor_sigma == 0, torch.full_like(factor_sigma, 1e-6), factor_sigma
        )

        mu = alpha_mu + torch.matmul(beta, factor_mu)
        sigma = torch.sqrt(alpha_sigma ** 2 + torch.matmul(beta ** 2, factor_sigma ** 2) + 1e-6)
        return self.reparameterize(mu, sigma)


class AttentionLayer(nn.Module):
    """Single-query cross-sectional attention head.

    Op order is the reference's and must be preserved: scores
    q·Kᵀ/sqrt(H+1e-6) -> Dropout(0.1) on PRE-softmax scores -> ReLU ->
    softmax over stocks (dim=0) -> NaN/Inf guard (zero context) ->
    context aᵀV. Reference: /root/reference/module.py:125-153.
    """

    def __init__(self, hidden_size: int):
        super().__init__()
        self.query = nn.Parameter(torch.randn(hidden_size))
        self.key_layer = nn.Linear(hidden_size, hidden_size)
        self.value_layer = nn.Linear(hidden_size, hidden_size)
        self.dropout = nn.Dropout(0.1)

    def forward(self, stock_latent: torch.Tensor) -> torch.Tensor:
        key = self.key_layer(stock_latent)      # (N, H)
        value = self.value_layer(stock_latent)  # (N, H)

        attention_weights = torch.matmul(self.query, key.transpose(1, 0))  # (N,)
        scale = torch.sqrt(torch.tensor(key.shape[1], dtype=torch.get_default_dtype(),
                                        device=key.device) + 1e-6)
        attention_weights = attention_weights / scale
        attention_weights = self.dropout(attention_weights)
        attention_weights = F.relu(attention_weights)
        attention_weights = F.softmax(attention_weights, dim=0)

        if torch.isnan(attention_weights).any() or torch.isinf(attention_weights).any():
            return torch.zeros_like(value[0])
        return torch.matmul(attention_weights, value)  # (H,)


class FactorPredictor(nn.Module):
    """Prior p(z|x): K independent attention heads -> (K,H) -> shared MLP
    heads. Reference: /root/reference/module.py:155-188.

    The eager path vectorizes the K heads (stacked weights, one batched
    pass) — numerically identical to the reference's Python loop, without
    its 2K serialized kernel launches. `attention_layers` keeps the
    reference's per-head parameter namespace for checkpoint parity.
    """

    def __init__(self, hidden_size: int, num_factor: int):
        super().__init__()
        self.hidden_size = hidden_size
        self.num_factor = num_factor
        self.attention_layers = nn.ModuleList(
            [AttentionLayer(hidden_size) for _ in range(num_factor)]
        )

        self.linear = nn.Linear(hidden_size, hidden_size)
        self.leakyrelu = nn.LeakyReLU()
        self.mu_layer = nn.Linear(hidden_size, 1)
        self.sigma_layer = nn.Linear(hidden_size, 1)
        self.softplus = nn.Softplus()

    def _heads_vectorized(self, stock_latent: torch.Tensor) -> torch.Tensor:
        """All K heads in one batched pass.

        Uses the identity q·(h Wkᵀ + bk)ᵀ = h (Wkᵀ q) + q·bk so per-head
        key matrices are never materialized, and context
        aᵀ(h Wvᵀ + bv) = Wv (aᵀ h) + (Σa) bv with Σa = 1 after softmax —
        the same restructure the HIP kernels use.
        """
        K, H = self.num_factor, self.hidden_size
        h = stock_latent  # (N, H)
        q = torch.stack([l.query for l in self.attention_layers])            # (K, H)
        Wk = torch.stack([l.key_layer.weight for l in self.attention_layers])   # (K, H, H)
        bk = torch.stack([l.key_layer.bias for l in self.attention_layers])     # (K, H)
        Wv = torch.stack([l.value_layer.weight for l in self.attention_layers])  # (K, H, H)
        bv = torch.stack([l.value_layer.bias for l in self.attention_layers])    # (K, H)

        # scores[n,k] = q_k · (Wk_k h_n + bk_k) = h_n · (Wk_kᵀ q_k) + q_k·bk_k
        qk = torch.einsum("khj,kh->kj", Wk, q)          # (K, H)  Wkᵀ q
        c = (q * bk).sum(dim=1)                          # (K,)
        scale = torch.sqrt(torch.tensor(H, dtype=h.dtype, device=h.device) + 1e-6)
        scores = (h @ qk.t() + c) / scale                # (N, K)

        if self.training:
            scores = F.dropout(scores, p=0.1, training=True)
        scores = F.relu(scores)
        a = F.softmax(scores, dim=0)                     # (N, K) softmax over stocks

        # NaN/Inf guard per head (reference returns a zero context vector)
        bad = torch.isnan(a).any(dim=0) | torch.isinf(a).any(dim=0)  # (K,)

        u = a.t() @ h                                    # (K, H)  aᵀ h
        ctx = torch.einsum("kij,kj->ki", Wv, u) + bv     # (K, H)
        ctx = torch.where(bad.unsqueeze(1), torch.zeros_like(ctx), ctx)
        return ctx

    def forward(self, stock_latent: torch.Tensor):
        h_multi = self._heads_vectorized(stock_latent)   # (K, H)
        h_multi = self.leakyrelu(self.linear(h_multi))
        pred_mu = self.mu_layer(h_multi).view(-1)
        pred_sigma = self.softplus(self.sigma_layer(h_multi)).view(-1)
        return pred_mu, pred_sigma

    def forward_looped(self, stock_latent: torch.Tensor):
        """Reference-shaped per-head loop (kept for parity testing only)."""
        outs = [layer(stock_latent) for layer in self.attention_layers]
        h_multi = torch.stack(outs, dim=0).view(self.num_factor, -1)
        h_multi = self.leakyrelu(self.linear(h_multi))
        pred_mu = self.mu_layer(h_multi).view(-1)
        pred_sigma = self.softplus(self.sigma_layer(h_multi)).view(-1)
        return pred_mu, pred_sigma


class FactorVAE(nn.Module):
    """Composition root. forward -> (loss, recon, mu_post, sigma_post,
    mu_prior, sigma_prior); prediction(x) -> (N,1) using the PRIOR.

    Reference: /root/reference/module.py:234-278.
    """

    def __init__(self, feature_extractor, factor_encoder, factor_decoder, factor_predictor):
        super().__init__()
        self.feature_extractor = feature_extractor
        self.factor_encoder = factor_encoder
        self.factor_decoder = factor_decoder
        self.factor_predictor = factor_predictor

    @staticmethod
    def KL_Divergence(mu1, sigma1, mu2, sigma2):
        # sum-reduced KL of diagonal Gaussians (reference module.py:242-248)
        return (torch.log(sigma2 / sigma1)
                + (sigma1 ** 2 + (mu1 - mu2) ** 2) / (2 * sigma2 ** 2) - 0.5).sum()

    def forward(self, x: torch.Tensor, returns: torch.Tensor):
        if returns.dim() == 1:  # accept (N,) like the encoder does; the
            returns = returns.unsqueeze(1)  # mse would silently broadcast (N,N)
        stock_latent = self.feature_extractor(x)
        factor_mu, factor_sigma = self.factor_encoder(stock_latent, returns)
        reconstruction = self.factor_decoder(stock_latent, factor_mu, factor_sigma)
        pred_mu, pred_sigma = self.factor_predictor(stock_latent)

        reconstruction_loss = F.mse_loss(reconstruction, returns)
        pred_sigma = torch.where(
            pred_sigma == 0, torch.full_like(pred_sigma, 1e-6), pred_sigma
        )
        kl_divergence = self.KL_Divergence(factor_mu, factor_sigma, pred_mu, pred_sigma)
        vae_loss = reconstruction_loss + kl_divergence
        return vae_loss, reconstruction, factor_mu, factor_sigma, pred_mu, pred_sigma

    @torch.no_grad()
    def prediction(self, x: torch.Tensor) -> torch.Tensor:
        stock_latent = self.feature_extractor(x)
        pred_mu, pred_sigma = self.factor_predictor(stock_latent)
        return self.factor_decoder(stock_latent, pred_mu, pred_sigma)


def build_factorvae(num_latent: int = 158, hidden_size: int = 64,
                    num_portfolio: int = 128, num_factor: int = 96) -> FactorVAE:
    """Construct the 6-module FactorVAE the way the reference CLI does
    (/root/reference/main.py:27-33)."""
    feature_extractor = FeatureExtractor(num_latent=num_latent, hidden_size=hidden_size)
    factor_encoder = FactorEncoder(num_factors=num_factor, num_portfolio=num_portfolio,
                                   hidden_size=hidden_size)
    alpha_layer = AlphaLayer(hidden_size)
    beta_layer = BetaLayer(hidden_size, num_factor)
    factor_decoder = FactorDecoder(alpha_layer, beta_layer)
    factor_predictor = FactorPredictor(hidden_size, num_factor)
    return FactorVAE(feature_extractor, factor_encoder, factor_decoder, factor_predictor)
