"""Data layer tests: synthetic generator, windowed sampler, day batching."""

import numpy as np
import pandas as pd
import pytest
import torch

from factorvae_amd.data.sampler import (
    DateGroupedBatchSampler,
    TSDataSampler,
    TSDatasetH,
    init_data_loader,
    np_ffill,
)
from factorvae_amd.data.synthetic import make_synthetic_frame


def test_np_ffill():
    a = np.array([np.nan, 1.0, np.nan, 3.0, np.nan])
    out = np_ffill(a)
    assert np.isnan(out[0])
    assert out[1] == 1.0 and out[2] == 1.0 and out[3] == 3.0 and out[4] == 3.0


def test_synthetic_frame_contract():
    df = make_synthetic_frame(n_days=10, n_stocks=7, n_features=12, seed=0)
    assert list(df.index.names) == ["datetime", "instrument"]
    assert df.shape == (70, 13)
    assert df.columns[-1] == "LABEL0"
    # label is CS-rank-normalized per day: zero-mean-ish, bounded
    day0 = df.loc[df.index.get_level_values(0)[0]]
    assert abs(day0["LABEL0"].mean()) < 0.5
    assert day0["LABEL0"].abs().max() <= np.sqrt(12.0) / 2 + 1e-5


def test_synthetic_deterministic():
    a = make_synthetic_frame(n_days=4, n_stocks=5, n_features=6, seed=3)
    b = make_synthetic_frame(n_days=4, n_stocks=5, n_features=6, seed=3)
    pd.testing.assert_frame_equal(a, b)


@pytest.fixture
def frame():
    return make_synthetic_frame(n_days=30, n_stocks=8, n_features=6, seed=1)


def test_sampler_window_shapes(frame):
    T = 5
    s = TSDataSampler(frame, start=frame.index.levels[0][0],
                      end=frame.index.levels[0][-1], step_len=T,
                      fillna_type="ffill+bfill")
    data, idx = s[0]
    assert data.shape == (T, 7)
    data, idx = s[[0, 1, 2]]
    assert data.shape == (3, T, 7)


def test_sampler_window_content(frame):
    """Window for day i, stock j = rows of that stock over days i-T+1..i."""
    T = 4
    dates = frame.index.levels[0]
    s = TSDataSampler(frame, start=dates[0], end=dates[-1], step_len=T)
    # pick a position well inside
    index = s.get_index()
    pos = 10 * 8  # day 10, first stock
    date, inst = index[pos]
    data, _ = s[pos]
    di = list(dates).index(date)
    for k in range(T):
        expect = frame.loc[(dates[di - T + 1 + k], inst)].to_numpy()
        assert np.allclose(data[k], expect, atol=1e-6)


def test_sampler_left_pad_bfill(frame):
    """First day's window: T-1 leading slots backfilled with the first row."""
    T = 6
    dates = frame.index.levels[0]
    s = TSDataSampler(frame, start=dates[0], end=dates[-1], step_len=T,
                      fillna_type="ffill+bfill")
    data, _ = s[0]
    for k in range(T - 1):
        assert np.allclose(data[k], data[T - 1], atol=1e-6)


def test_sampler_left_pad_nan_mode(frame):
    T = 6
    dates = frame.index.levels[0]
    s = TSDataSampler(frame, start=dates[0], end=dates[-1], step_len=T,
                      fillna_type="none")
    data, _ = s[0]
    assert np.isnan(data[:T - 1]).all()
    assert np.isfinite(data[T - 1]).all()


def test_sampler_gap_ffill():
    """A stock missing on a middle day: window-local ffill fills the gap
    with the previous present row (reference dataset.py:139-151)."""
    df = make_synthetic_frame(n_days=10, n_stocks=4, n_features=3, seed=2)
    dates = df.index.levels[0]
    # drop stock 0 on day 5
    inst0 = df.index.levels[1][0]
    df2 = df.drop(index=(dates[5], inst0))
    s = TSDataSampler(df2, start=dates[0], end=dates[-1], step_len=4,
                      fillna_type="ffill+bfill")
    index = s.get_index()
    # find position of (day 7, inst0): window covers days 4,5,6,7; day-5 slot
    pos = index.get_loc((dates[7], inst0))
    data, _ = s[pos]
    day4 = df2.loc[(dates[4], inst0)].to_numpy()
    assert np.allclose(data[0], day4, atol=1e-6)
    assert np.allclose(data[1], day4, atol=1e-6)  # gap ffilled from day 4
    assert np.allclose(data[2], df2.loc[(dates[6], inst0)].to_numpy(), atol=1e-6)


def test_date_grouped_batches(frame):
    loader = init_data_loader(frame, step_len=5, shuffle=False,
                              start=frame.index.levels[0][0],
                              end=frame.index.levels[0][-1])
    batches = list(loader)
    assert len(batches) == 30  # one batch per trading day
    for data, idx in batches:
        assert data.shape == (8, 5, 7)  # all 8 stocks of the day
        assert isinstance(data, torch.Tensor)


def test_date_grouped_batches_ragged():
    df = make_synthetic_frame(n_days=12, n_stocks=20, n_features=4, seed=5, ragged=True)
    loader = init_data_loader(df, step_len=3, shuffle=False,
                              start=df.index.levels[0][0], end=df.index.levels[0][-1])
    sizes = [data.shape[0] for data, _ in loader]
    assert len(sizes) == 12
    assert min(sizes) >= 5 and max(sizes) <= 20
    assert len(set(sizes)) > 1  # genuinely variable N per day


def test_day_sharding_covers_all_days(frame):
    ds = TSDatasetH(frame, step_len=5, start=frame.index.levels[0][0],
                    end=frame.index.levels[0][-1], fillna_type="ffill+bfill")
    all_days = set()
    for rank in range(3):
        s = DateGroupedBatchSampler(ds, shuffle=True, rank=rank, world_size=3, seed=7)
        s.set_epoch(2)
        for group in s:
            all_days.add(tuple(group))
    # disjoint union over ranks covers every day exactly once
    assert sum(len(list(DateGroupedBatchSampler(ds, shuffle=True, rank=r,
                                                world_size=3, seed=7)))
               for r in range(3)) == 30
    assert len(all_days) == 30


def test_shuffle_is_epoch_seeded(frame):
    ds = TSDatasetH(frame, step_len=5, start=frame.index.levels[0][0],
                    end=frame.index.levels[0][-1], fillna_type="ffill+bfill")
    s = DateGroupedBatchSampler(ds, shuffle=True, seed=3)
    s.set_epoch(0)
    o1 = [g[0] for g in s]
    s.set_epoch(0)
    o2 = [g[0] for g in s]
    s.set_epoch(1)
    o3 = [g[0] for g in s]
    assert o1 == o2
    assert o1 != o3


def test_make_dataset_cli(tmp_path):
    """Reference-layout ETL entrypoint (data/make_dataset.py) writes a
    loadable dataset pickle with the reference output contract."""
    import os
    import subprocess
    import sys

    import pandas as pd

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = tmp_path / "mk.pkl"
    r = subprocess.run(
        [sys.executable, os.path.join(root, "data", "make_dataset.py"),
         "--out", str(out), "--n_days", "10", "--n_stocks", "6"],
        capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, r.stderr[-1500:]
    df = pd.read_pickle(out)
    assert list(df.index.names) == ["datetime", "instrument"]
    assert df.shape == (60, 159) and "LABEL0" in df.columns
