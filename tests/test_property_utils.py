"""Property-based tests of numeric utilities (CPU).

np_ffill against pandas' ffill, and RankIC against a direct per-day
Spearman computation — independent implementations of the same specs
(/root/reference/utils.py:113-129, dataset.py:24-39). Derandomized for
run-to-run determinism.
"""

import numpy as np
import pandas as pd
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st
from scipy.stats import spearmanr

from factorvae_amd.utils import RankIC
from factorvae_amd.data.sampler import np_ffill

SET = dict(derandomize=True, max_examples=60, deadline=None)


@given(
    rows=st.integers(1, 12),
    cols=st.integers(0, 4),  # 0 -> 1-D input
    nan_bits=st.integers(0, 2**48 - 1),
    seed=st.integers(0, 10**6),
)
@settings(**SET)
def test_np_ffill_matches_pandas(rows, cols, nan_bits, seed):
    """This test found a real bug: the reference-shaped np_ffill only
    broadcast correctly for 1-D input (silently wrong for square 2-D,
    ValueError otherwise); the 2-D per-column path was added for it."""
    rng = np.random.default_rng(seed)
    a = rng.standard_normal((rows, cols) if cols else (rows,))
    flat = a.reshape(-1)
    for b in range(flat.size):
        if (nan_bits >> (b % 48)) & 1:
            flat[b] = np.nan
    a = flat.reshape(rows, cols) if cols else flat
    got = np_ffill(a.copy())
    exp = (pd.DataFrame(a) if cols else pd.Series(a)).ffill().to_numpy()
    np.testing.assert_array_equal(got, exp)


@given(
    n_days=st.integers(1, 6),
    n_stocks=st.integers(3, 20),
    noise=st.floats(0.0, 2.0),
    seed=st.integers(0, 10**6),
)
@settings(**SET)
def test_rankic_matches_direct_spearman(n_days, n_stocks, noise, seed):
    rng = np.random.default_rng(seed)
    dates = pd.date_range("2022-01-03", periods=n_days, freq="B")
    idx = pd.MultiIndex.from_product(
        [dates, [f"S{i}" for i in range(n_stocks)]],
        names=["datetime", "instrument"])
    label = rng.standard_normal(len(idx))
    pred = label + noise * rng.standard_normal(len(idx))
    df = pd.DataFrame({"LABEL0": label, "Pred": pred}, index=idx)

    out = RankIC(df)
    assert isinstance(out, pd.DataFrame)

    daily = [spearmanr(df.loc[d, "LABEL0"], df.loc[d, "Pred"])[0]
             for d in dates]
    exp_ric = float(np.mean(daily))
    np.testing.assert_allclose(out["RankIC"].iloc[0], exp_ric, atol=1e-12)
    std = float(np.std(daily))
    if std != 0:
        np.testing.assert_allclose(out["RankIC_IR"].iloc[0], exp_ric / std,
                                   atol=1e-9)
    # perfect monotone prediction -> RankIC exactly 1
    df2 = pd.DataFrame({"LABEL0": label, "Pred": 3.0 * label + 1.0},
                       index=idx)
    np.testing.assert_allclose(RankIC(df2)["RankIC"].iloc[0], 1.0)


def test_rankic_empty_frame_returns_dataframe():
    idx = pd.MultiIndex.from_arrays(
        [pd.DatetimeIndex([]), pd.Index([])],
        names=["datetime", "instrument"])
    out = RankIC(pd.DataFrame({"LABEL0": [], "Pred": []}, index=idx))
    assert isinstance(out, pd.DataFrame) and np.isnan(out["RankIC"].iloc[0])


# ---------------------------------------------------------------------------
# risk_analysis differential: brute-force formulas on random return
# series, plus structural invariants (drawdown <= 0, scale equivariance).
# ---------------------------------------------------------------------------

from factorvae_amd.backtest import risk_analysis


@given(
    n=st.integers(2, 60),
    seed=st.integers(0, 10**6),
    mu=st.floats(-0.01, 0.01),
    vol=st.floats(1e-4, 0.05),
)
@settings(**SET)
def test_risk_analysis_matches_brute_force(n, seed, mu, vol):
    rng = np.random.default_rng(seed)
    r = pd.Series(mu + vol * rng.standard_normal(n),
                  index=pd.date_range("2023-01-02", periods=n, freq="B"))
    out = risk_analysis(r, N=252)["risk"]
    assert list(out.index) == ["mean", "std", "annualized_return",
                               "information_ratio", "max_drawdown"]
    np.testing.assert_allclose(out["mean"], r.to_numpy().mean(), atol=1e-12)
    np.testing.assert_allclose(out["std"], np.std(r.to_numpy(), ddof=1),
                               rtol=1e-10)
    np.testing.assert_allclose(out["annualized_return"], out["mean"] * 252,
                               rtol=1e-12)
    np.testing.assert_allclose(out["information_ratio"],
                               out["mean"] / out["std"] * np.sqrt(252),
                               rtol=1e-10)
    # max drawdown: brute-force over all (i <= j) windows of the cumsum
    cum = np.cumsum(r.to_numpy())
    dd = min(cum[j] - cum[:j + 1].max() for j in range(n))
    np.testing.assert_allclose(out["max_drawdown"], dd, atol=1e-12)
    assert out["max_drawdown"] <= 0
    # NaNs are dropped, not poisoning
    r2 = r.copy()
    r2.iloc[0] = np.nan
    assert np.isfinite(risk_analysis(r2, N=252)["risk"]["mean"])


# ---------------------------------------------------------------------------
# Top-k dropout simulator invariants across random score/return panels.
# ---------------------------------------------------------------------------

from factorvae_amd.backtest import BacktestConfig, topk_dropout_backtest


@given(
    n_days=st.integers(1, 8),
    n_stocks=st.integers(1, 15),
    topk=st.integers(1, 8),
    n_drop=st.integers(0, 4),
    seed=st.integers(0, 10**6),
)
@settings(**SET)
def test_topk_dropout_invariants(n_days, n_stocks, topk, n_drop, seed):
    rng = np.random.default_rng(seed)
    dates = pd.date_range("2023-03-01", periods=n_days, freq="B")
    idx = pd.MultiIndex.from_product(
        [dates, [f"S{i}" for i in range(n_stocks)]],
        names=["datetime", "instrument"])
    df = pd.DataFrame({
        "score": rng.standard_normal(len(idx)),
        "LABEL0": 0.02 * rng.standard_normal(len(idx)),
    }, index=idx)
    cfg = BacktestConfig(topk=topk, n_drop=n_drop)
    res = topk_dropout_backtest(df, config=cfg)

    assert len(res.holdings) == n_days
    held_prev = []
    for d, held in enumerate(res.holdings):
        day_scores = df.loc[dates[d], "score"]
        # book size: never above topk; equals min(topk, universe) when
        # enough candidates exist
        assert len(held) <= topk
        assert len(set(held)) == len(held)  # no duplicate names
        assert set(held) <= set(day_scores.index)
        if d == 0:
            assert held == list(
                day_scores.sort_values(ascending=False).index[:topk])
        else:
            # at most n_drop names leave the book per day
            dropped = set(held_prev) - set(held)
            assert len(dropped) <= n_drop
        held_prev = held

    # turnover/cost bounds: first day all buys; after that at most
    # (n_drop sells + n_drop buys)/k of the book
    assert (res.daily_turnover >= 0).all()
    assert (res.daily_cost >= 0).all()
    assert res.daily_turnover.iloc[0] <= 1.0 + 1e-12
    if n_days > 1:
        k = max(1, min(topk, n_stocks))
        assert (res.daily_turnover.iloc[1:] <= 2 * n_drop / k + 1e-12).all()

    # returns: equal-weight mean of held names' LABEL0
    for d, held in enumerate(res.holdings):
        exp = (df.loc[dates[d], "LABEL0"].reindex(held).fillna(0).mean()
               if held else 0.0)
        np.testing.assert_allclose(res.daily_return.iloc[d], exp, atol=1e-12)

    # with-cost excess = no-cost excess - cost (no benchmark given)
    np.testing.assert_allclose(
        res.excess_with_cost.to_numpy(),
        (res.excess_no_cost - res.daily_cost).to_numpy(), atol=1e-15)


# ---------------------------------------------------------------------------
# Synthetic generator invariants — it underpins most other tests.
# ---------------------------------------------------------------------------

from factorvae_amd.data.synthetic import make_synthetic_frame


@given(
    n_days=st.integers(1, 10),
    n_stocks=st.integers(5, 40),
    n_features=st.integers(1, 12),
    seed=st.integers(0, 10**6),
    ragged=st.booleans(),
    lff=st.booleans(),
)
@settings(**SET)
def test_synthetic_frame_invariants(n_days, n_stocks, n_features, seed,
                                    ragged, lff):
    df = make_synthetic_frame(n_days=n_days, n_stocks=n_stocks,
                              n_features=n_features, seed=seed,
                              ragged=ragged, label_from_features=lff)
    assert list(df.index.names) == ["datetime", "instrument"]
    assert df.index.is_monotonic_increasing  # sorted as loaders expect
    assert not df.index.duplicated().any()
    assert list(df.columns[-1:]) == ["LABEL0"]
    assert df.shape[1] == n_features + 1
    assert np.isfinite(df.to_numpy()).all()  # no NaN/Inf anywhere
    assert df.dtypes.eq(np.float32).all()
    sizes = df.groupby(level=0).size()
    assert len(sizes) == n_days
    if ragged and n_stocks > 10:
        assert (sizes >= 5).all() and (sizes <= n_stocks).all()
    else:
        assert (sizes == n_stocks).all()
    # CSRankNorm label: bounded by ±0.5*sqrt(12); pct-rank gives an
    # exact per-day mean of sqrt(12)/(2N) when there are no ties
    lab = df["LABEL0"]
    bound = 0.5 * np.sqrt(12.0) + 1e-5
    assert (lab.abs() <= bound).all()
    for _, day in lab.groupby(level=0):
        np.testing.assert_allclose(day.mean(),
                                   np.sqrt(12.0) / (2 * len(day)),
                                   atol=1e-3)
    # determinism: same seed -> identical frame
    df2 = make_synthetic_frame(n_days=n_days, n_stocks=n_stocks,
                               n_features=n_features, seed=seed,
                               ragged=ragged, label_from_features=lff)
    pd.testing.assert_frame_equal(df, df2)
