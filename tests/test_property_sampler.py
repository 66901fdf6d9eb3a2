"""Property-based differential tests of TSDataSampler (CPU).

The sampler is a from-scratch algorithm (integer-factorized id matrix +
vectorized window gather, factorvae_amd/data/sampler.py) implementing
the reference's qlib-style semantics (/root/reference/dataset.py:41-274:
trailing T-window per (date, instrument), missing rows resolved
window-locally by none/ffill/ffill+bfill). Here we check it against an
independent pandas implementation of the same spec — Series.unstack for
the id matrix and Series.ffill/bfill for the window fill — across
hypothesis-generated frames with random presence patterns.

`derandomize=True` keeps the suite deterministic run-to-run.
"""

import numpy as np
import pandas as pd
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from factorvae_amd.data.sampler import TSDataSampler

SET = dict(derandomize=True, max_examples=60, deadline=None)


def make_frame(n_dates, n_inst, present_bits, nan_bits, n_cols=2):
    """Random-presence (datetime, instrument) frame; values encode their
    (date, inst, col) identity so gather errors are visible; some values
    are NaN (in-row NaNs must pass through untouched — fill semantics
    operate on MISSING ROWS, not NaN values)."""
    dates = pd.date_range("2020-01-01", periods=n_dates, freq="D")
    insts = [f"S{i}" for i in range(n_inst)]
    rows, vals = [], []
    b = 0
    for d in range(n_dates):
        for u in range(n_inst):
            if (present_bits >> b) & 1:
                rows.append((dates[d], insts[u]))
                v = [d * 100.0 + u * 10.0 + c for c in range(n_cols)]
                if (nan_bits >> b) & 1:
                    v[0] = np.nan
                vals.append(v)
            b += 1
    if not rows:
        return None
    idx = pd.MultiIndex.from_tuples(rows, names=["datetime", "instrument"])
    return pd.DataFrame(vals, index=idx,
                        columns=[f"f{c}" for c in range(n_cols)])


def oracle_window(df, date, inst, T, fillna_type):
    """Independent spec implementation via pandas unstack + Series fill."""
    ids = pd.Series(np.arange(len(df), dtype=float), index=df.index)
    mat = ids.unstack(level="instrument")          # (dates x insts), NaN absent
    di = mat.index.get_loc(date)
    col = mat[inst].to_numpy()
    lo = di - T + 1
    win = np.full(T, np.nan)
    src_lo = max(lo, 0)
    win[src_lo - lo:] = col[src_lo:di + 1]
    s = pd.Series(win)
    if fillna_type in ("ffill", "ffill+bfill"):
        s = s.ffill()
    if fillna_type == "ffill+bfill":
        s = s.bfill()
    w = s.to_numpy()
    arr = df.to_numpy(dtype=np.float32)
    out = np.full((T, arr.shape[1]), np.nan, dtype=np.float32)
    ok = ~np.isnan(w)
    out[ok] = arr[w[ok].astype(int)]
    return out


@given(
    n_dates=st.integers(2, 7),
    n_inst=st.integers(1, 4),
    present_bits=st.integers(0, 2**28 - 1),
    nan_bits=st.integers(0, 2**28 - 1),
    T=st.integers(1, 5),
    fillna=st.sampled_from(["none", "ffill", "ffill+bfill"]),
)
@settings(**SET)
def test_window_matches_pandas_oracle(n_dates, n_inst, present_bits,
                                      nan_bits, T, fillna):
    df = make_frame(n_dates, n_inst, present_bits, nan_bits)
    if df is None:
        return
    smp = TSDataSampler(df, None, None, step_len=T, fillna_type=fillna)
    assert len(smp) == len(df)
    for p in range(len(smp)):
        got, actual = smp[p]
        date, inst = actual[0]
        exp = oracle_window(df.sort_index(), date, inst, T, fillna)
        np.testing.assert_array_equal(
            got, exp,
            err_msg=f"(date={date.date()}, inst={inst}, T={T}, {fillna})")


@given(
    n_dates=st.integers(2, 7),
    n_inst=st.integers(1, 4),
    present_bits=st.integers(0, 2**28 - 1),
    T=st.integers(1, 4),
)
@settings(**SET)
def test_batch_path_equals_item_path(n_dates, n_inst, present_bits, T):
    """The vectorized list-__getitem__ must agree with the scalar path."""
    df = make_frame(n_dates, n_inst, present_bits, 0)
    if df is None:
        return
    smp = TSDataSampler(df, None, None, step_len=T, fillna_type="ffill")
    all_idx = list(range(len(smp)))
    batch, actual = smp[all_idx]
    assert batch.shape == (len(smp), T, df.shape[1])
    for p in all_idx:
        one, a1 = smp[p]
        np.testing.assert_array_equal(batch[p], one)
        assert actual[p] == a1[0]


@given(
    n_dates=st.integers(3, 8),
    n_inst=st.integers(1, 3),
    present_bits=st.integers(1, 2**24 - 1),
    lo_frac=st.floats(0.0, 1.0),
    hi_frac=st.floats(0.0, 1.0),
)
@settings(**SET)
def test_start_end_slicing_matches_index(n_dates, n_inst, present_bits,
                                         lo_frac, hi_frac):
    """start/end date slicing: get_index() is exactly the full sorted
    index restricted to [start, end] and positions map accordingly."""
    df = make_frame(n_dates, n_inst, present_bits, 0)
    if df is None:
        return
    dates = df.index.get_level_values(0).unique().sort_values()
    lo = dates[int(lo_frac * (len(dates) - 1))]
    hi = dates[int(hi_frac * (len(dates) - 1))]
    if lo > hi:
        lo, hi = hi, lo
    smp = TSDataSampler(df, lo, hi, step_len=2)
    full = df.sort_index().index
    expect = full[(full.get_level_values(0) >= lo)
                  & (full.get_level_values(0) <= hi)]
    assert smp.get_index().equals(expect)
    assert len(smp) == len(expect)
    if len(smp):
        _, actual = smp[0]
        assert actual[0] == expect[0]
        _, actual = smp[len(smp) - 1]
        assert actual[0] == expect[-1]


# ---------------------------------------------------------------------------
# DP day-sharding invariants (DateGroupedBatchSampler): these properties
# are what keeps the driver's N-GPU run collectively consistent — equal
# per-rank step counts (equal all-reduce counts), full day coverage, and
# an epoch-synchronized shuffle.
# ---------------------------------------------------------------------------

from factorvae_amd.data.sampler import DateGroupedBatchSampler


class _Src:
    class _S:
        def __init__(self, index):
            self._i = index

        def get_index(self):
            return self._i

    def __init__(self, n_days, insts_per_day):
        dates = pd.date_range("2021-01-04", periods=n_days, freq="B")
        tup = [(d, f"S{u}") for i, d in enumerate(dates)
               for u in range(insts_per_day[i])]
        self.sampler = self._S(pd.MultiIndex.from_tuples(
            tup, names=["datetime", "instrument"]))


@given(
    n_days=st.integers(1, 13),
    world_size=st.integers(1, 8),
    shuffle=st.booleans(),
    epoch=st.integers(0, 3),
    sizes=st.lists(st.integers(1, 4), min_size=13, max_size=13),
)
@settings(**SET)
def test_dp_shard_invariants(n_days, world_size, shuffle, epoch, sizes):
    src = _Src(n_days, sizes)
    samplers = []
    for r in range(world_size):
        s = DateGroupedBatchSampler(src, shuffle=shuffle, rank=r,
                                    world_size=world_size, seed=5)
        s.set_epoch(epoch)
        samplers.append(s)
    per_rank = [list(s) for s in samplers]

    # 1. equal step counts on every rank == len(sampler)
    want = -(-n_days // world_size)  # ceil
    assert all(len(b) == want == len(s)
               for b, s in zip(per_rank, samplers))

    # 2. batches are intact whole days (exactly one date per batch)
    index = src.sampler.get_index()
    days_of = lambda batch: {index[i][0] for i in batch}
    for batches in per_rank:
        for b in batches:
            assert len(days_of(b)) == 1

    # 3. full coverage: every day appears on some rank; and when no
    #    padding is needed, the shard is an exact partition
    seen = [d for batches in per_rank for b in batches for d in days_of(b)]
    assert len(set(seen)) == n_days
    if n_days % world_size == 0:
        assert len(seen) == n_days  # each day exactly once globally

    # 4. ranks agree on one global order: round-robin interleave of the
    #    rank streams is a single consistent sequence (first `n_days`
    #    entries cover all days; padding only repeats earlier days)
    interleaved = []
    for k in range(want):
        for r in range(world_size):
            if k < len(per_rank[r]):
                interleaved.append(next(iter(days_of(per_rank[r][k]))))
    assert len(set(interleaved[:n_days])) == n_days
    assert set(interleaved[n_days:]) <= set(interleaved[:n_days])

    # 5. same epoch -> identical order on a fresh sampler (determinism)
    s2 = DateGroupedBatchSampler(src, shuffle=shuffle, rank=0,
                                 world_size=world_size, seed=5)
    s2.set_epoch(epoch)
    assert list(s2) == per_rank[0]


@given(
    n_dates=st.integers(2, 6),
    n_inst=st.integers(2, 4),
    flt_bits=st.integers(0, 2**24 - 1),
    T=st.integers(1, 4),
)
@settings(**SET)
def test_flt_data_row_filter(n_dates, n_inst, flt_bits, T):
    """flt_data restricts WHICH rows are sampled, but windows still
    gather history from the FULL frame (the reference's fltdata
    semantics): index/len reflect the filtered rows, window content
    matches the unfiltered sampler at the same (date, inst)."""
    df = make_frame(n_dates, n_inst, 2**28 - 1, 0)  # full presence
    flt = pd.Series(
        [(flt_bits >> (i % 24)) & 1 == 1 for i in range(len(df))],
        index=df.index)
    if not flt.any():
        return
    smp = TSDataSampler(df, None, None, step_len=T, fillna_type="ffill",
                        flt_data=flt)
    full = TSDataSampler(df, None, None, step_len=T, fillna_type="ffill")
    expect_index = df.sort_index().index[flt.to_numpy()]
    assert smp.get_index().equals(expect_index)
    assert len(smp) == int(flt.sum())
    # each filtered position's window equals the unfiltered sampler's
    # window at the corresponding full-frame position
    full_pos = {key: p for p, key in enumerate(df.sort_index().index)}
    for p in range(len(smp)):
        got, actual = smp[p]
        exp, _ = full[full_pos[actual[0]]]
        np.testing.assert_array_equal(got, exp)
