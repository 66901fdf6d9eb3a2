"""Synthetic Alpha158-shaped data generator.

Replicates the *output contract* of the reference ETL
(/root/reference/data/make_dataset.py:66-83): a pandas DataFrame with
MultiIndex (datetime, instrument), 158 feature columns (RobustZScore-like
scale) and a LABEL0 column (CSRankNorm-like daily cross-sectional rank
normalization of next-period returns). No qlib, no network: data is
generated from a seeded latent-factor model so that RankIC-style quality
metrics are non-trivial (labels correlate with a low-rank function of
features).
"""

from __future__ import annotations

import numpy as np
import pandas as pd

N_ALPHA_FEATURES = 158


def make_synthetic_frame(
    n_days: int = 60,
    n_stocks: int = 50,
    n_features: int = N_ALPHA_FEATURES,
    seed: int = 0,
    start: str = "2015-01-01",
    ragged: bool = False,
    signal_rank: int = 8,
    signal_strength: float = 0.15,
    label_from_features: bool = False,
) -> pd.DataFrame:
    """Build a synthetic (datetime, instrument)-indexed frame.

    Features follow an AR(1)-in-time latent factor model per stock; the
    label is the CS-rank-normalized next-period "return" driven by a
    low-rank projection of the features plus noise, so a trained model
    can achieve positive RankIC.

    ragged=True drops a random ~10% of stocks on each day (variable N per
    day, like real universes).

    label_from_features=True makes the return a fixed hidden linear
    function of the day's OWN feature vector (plus noise) — i.e. a
    signal a cross-sectional model can actually learn from x_t, unlike
    the default latent-factor returns (whose day factor is zero-mean
    given the features). Use it for end-to-end learning tests/demos.
    """
    rng = np.random.default_rng(seed)
    dates = pd.bdate_range(start, periods=n_days)
    instruments = [f"SH{600000 + i:06d}" for i in range(n_stocks)]

    # latent factors: (n_days, signal_rank) AR(1)
    f = np.zeros((n_days, signal_rank), dtype=np.float64)
    for t in range(1, n_days):
        f[t] = 0.9 * f[t - 1] + 0.44 * rng.standard_normal(signal_rank)

    # per-stock loadings onto latent factors and onto features
    loadings = rng.standard_normal((n_stocks, signal_rank)) * 0.5
    feat_mix = rng.standard_normal((signal_rank, n_features)) * 0.3
    # hidden feature->return weights for the learnable-label mode
    w_hidden = rng.standard_normal(n_features) / np.sqrt(n_features)

    rows = []
    index = []
    for t, date in enumerate(dates):
        if ragged and n_stocks > 10:
            day_n = int(rng.integers(int(n_stocks * 0.8), n_stocks + 1))
            keep = np.sort(rng.choice(n_stocks, size=max(5, day_n), replace=False))
        else:
            keep = np.arange(n_stocks)

        # features: stock-specific noise + factor-driven common part
        common = (loadings[keep] * f[t]).sum(axis=1)
        feats = loadings[keep] @ feat_mix + rng.standard_normal((len(keep), n_features))
        feats = feats.astype(np.float32)

        # next-period "return": signal + idiosyncratic noise
        if label_from_features:
            raw_ret = (signal_strength * (feats @ w_hidden)
                       + 0.3 * rng.standard_normal(len(keep)))
        else:
            raw_ret = signal_strength * common + rng.standard_normal(len(keep))
        # CSRankNorm: rank -> centered/scaled (qlib convention: (rank-0.5)*sqrt(12))
        rank = pd.Series(raw_ret).rank(pct=True).to_numpy()
        label = ((rank - 0.5) * np.sqrt(12.0)).astype(np.float32)

        for idx_in_keep, s in enumerate(keep):
            index.append((date, instruments[s]))
            rows.append(np.concatenate([feats[idx_in_keep], label[idx_in_keep:idx_in_keep + 1]]))

    cols = [f"FEAT{i:03d}" for i in range(n_features)] + ["LABEL0"]
    df = pd.DataFrame(
        np.asarray(rows, dtype=np.float32),
        index=pd.MultiIndex.from_tuples(index, names=["datetime", "instrument"]),
        columns=cols,
    )
    return df


def main():
    import argparse

    p = argparse.ArgumentParser(description="Generate a synthetic Alpha158-shaped pickle "
                                            "(same contract as the reference ETL output)")
    p.add_argument("--out", type=str, default="./data/synthetic_data.pkl")
    p.add_argument("--n_days", type=int, default=500)
    p.add_argument("--n_stocks", type=int, default=300)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--ragged", action="store_true")
    p.add_argument("--start", type=str, default="2015-01-01")
    args = p.parse_args()

    import os

    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    df = make_synthetic_frame(n_days=args.n_days, n_stocks=args.n_stocks,
                              seed=args.seed, ragged=args.ragged, start=args.start)
    df.to_pickle(args.out)
    print(f"wrote {args.out}: {df.shape[0]} rows x {df.shape[1]} cols, "
          f"{df.index.get_level_values(0).nunique()} days")


if __name__ == "__main__":
    main()
