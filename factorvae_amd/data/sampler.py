"""Time-series sampling over (datetime, instrument) frames.

Same interface and semantics as the reference's qlib-style stack
(/root/reference/dataset.py:41-274): `TSDataSampler` (T-window per
(date, instrument) with left NaN padding and ffill/bfill on the row-index
level), `TSDatasetH`, `DateGroupedBatchSampler` (one whole trading day
per batch — the unit of data parallelism), `custom_collate_fn`,
`init_data_loader`. New implementation: the (date x instrument) row-id
matrix is built with integer factorization instead of an object-dtype
unstack, and window gathers are fully vectorized.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
import torch
from torch.utils.data import DataLoader, Dataset, Sampler


def np_ffill(arr: np.ndarray) -> np.ndarray:
    """Forward-fill NaN along axis 0 (/root/reference/dataset.py:24-39).

    The reference version broadcasts incorrectly for 2-D inputs (it only
    ever calls it on 1-D data); here the 2-D case fills each column
    independently, matching ``pd.DataFrame(arr).ffill()`` — leading NaNs
    stay NaN. 1-D semantics are identical to the reference.
    """
    mask = np.isnan(arr)
    rows = np.arange(mask.shape[0])
    if arr.ndim == 1:
        idx = np.where(~mask, rows, 0)
        np.maximum.accumulate(idx, axis=0, out=idx)
        return arr[idx]
    idx = np.where(~mask, rows[:, None], 0)
    np.maximum.accumulate(idx, axis=0, out=idx)
    return arr[idx, np.arange(arr.shape[1])[None, :]]


class TSDataSampler:
    """Windowed sampler over a sorted MultiIndex (datetime, instrument)
    frame.

    For positional index p in [start_idx, end_idx) (positions within the
    sorted full frame), returns the (step_len, n_cols) block of the
    trailing T-window of that (date, instrument), with missing
    (date, inst) rows resolved by `fillna_type`:
      - "none": NaN rows
      - "ffill": previous valid row of the same instrument
      - "ffill+bfill": ffill, then remaining leading gaps backfilled
    """

    def __init__(self, data: pd.DataFrame, start, end, step_len: int,
                 fillna_type: str = "none", dtype=None, flt_data=None):
        assert fillna_type in ("none", "ffill", "ffill+bfill")
        assert list(data.index.names) == ["datetime", "instrument"]
        self.start = start
        self.end = end
        self.step_len = step_len
        self.fillna_type = fillna_type

        self.data = data.sort_index()
        self.data_index = self.data.index

        arr = self.data.to_numpy(dtype=np.float32 if dtype is None else dtype)
        # trailing all-NaN sentinel row (index -1)
        self.data_arr = np.append(
            arr, np.full((1, arr.shape[1]), np.nan, dtype=arr.dtype), axis=0
        )
        self.nan_idx = -1

        # (n_dates x n_instruments) matrix of row positions, -1 where absent
        dt_codes = self.data_index.codes[0]
        inst_codes = self.data_index.codes[1]
        self.dates = self.data_index.levels[0]
        self.instruments = self.data_index.levels[1]
        n_dates, n_inst = len(self.dates), len(self.instruments)

        self.id_matrix = np.full((n_dates, n_inst), -1, dtype=np.int64)
        self.id_matrix[dt_codes, inst_codes] = np.arange(len(self.data_index))

        # per-row (date_code, inst_code)
        self.row_date = np.asarray(dt_codes, dtype=np.int64)
        self.row_inst = np.asarray(inst_codes, dtype=np.int64)

        if flt_data is not None:
            flt = flt_data.reindex(self.data_index).fillna(False).astype(bool).to_numpy()
            self.flt_rows = np.nonzero(flt)[0]
            self.data_index = self.data_index[flt]
        else:
            self.flt_rows = None

        self.start_idx, self.end_idx = self.data_index.slice_locs(
            start=pd.Timestamp(start) if start is not None else None,
            end=pd.Timestamp(end) if end is not None else None,
        )

    def get_index(self) -> pd.MultiIndex:
        return self.data_index[self.start_idx:self.end_idx]

    def __len__(self) -> int:
        return self.end_idx - self.start_idx

    def _resolve_pos(self, idx) -> int:
        """Positional idx within [0, len) -> row position in full frame."""
        real = self.start_idx + idx
        if not (self.start_idx <= real < self.end_idx):
            raise KeyError(f"{real} out of bounds [{self.start_idx}, {self.end_idx})")
        if self.flt_rows is not None:
            return int(self.flt_rows[real])
        return int(real)

    def _window_ids(self, rows: np.ndarray) -> np.ndarray:
        """(B,) full-frame row positions -> (B, T) gather ids (-1 = NaN row).

        Fill is WINDOW-LOCAL, matching the reference (ffill/bfill run on
        the T-length index slice, /root/reference/dataset.py:139-151):
        a gap is forward-filled only from rows inside the window, and
        leading gaps are backfilled from the window's first valid row.
        """
        T = self.step_len
        i = self.row_date[rows]          # (B,)
        j = self.row_inst[rows]          # (B,)
        toff = np.arange(T) - (T - 1)    # offsets i-T+1 .. i
        ti = i[:, None] + toff[None, :]  # (B, T)
        in_range = ti >= 0
        ids = self.id_matrix[np.clip(ti, 0, None), j[:, None]]
        ids = np.where(in_range, ids, -1)

        if self.fillna_type in ("ffill", "ffill+bfill"):
            for t in range(1, T):
                ids[:, t] = np.where(ids[:, t] < 0, ids[:, t - 1], ids[:, t])
        if self.fillna_type == "ffill+bfill":
            for t in range(T - 2, -1, -1):
                ids[:, t] = np.where(ids[:, t] < 0, ids[:, t + 1], ids[:, t])
        return ids

    def __getitem__(self, idx):
        if isinstance(idx, (list, np.ndarray)):
            pos = np.asarray(idx, dtype=np.int64) + self.start_idx
            if len(pos) and (pos.min() < self.start_idx or
                             pos.max() >= self.end_idx):
                raise KeyError("index out of bounds")
            rows = (self.flt_rows[pos] if self.flt_rows is not None
                    else pos)
            ids = self._window_ids(rows)
            data = self.data_arr[ids]                      # (B, T, C+1)
            actual = self.data.index[rows]
            return data, actual
        rows = np.asarray([self._resolve_pos(idx)], dtype=np.int64)
        ids = self._window_ids(rows)
        data = self.data_arr[ids][0]                       # (T, C+1)
        actual = self.data.index[rows]
        return data, actual


class TSDatasetH(Dataset):
    """Thin Dataset wrapper (/root/reference/dataset.py:187-204)."""

    DEFAULT_STEP_LEN = 20

    def __init__(self, data, step_len: int = DEFAULT_STEP_LEN, **kwargs):
        self.step_len = step_len
        self.data = data
        self.sampler = TSDataSampler(data=data, step_len=step_len, **kwargs)

    def __getitem__(self, idx):
        return self.sampler[idx]

    def __getitems__(self, indices):
        """torch DataLoader batched-fetch protocol: gather the whole
        day-batch with ONE vectorized sampler call (the per-sample path
        pays pandas MultiIndex costs per row — ~15 days/s; this path
        does ~500+ days/s). Returns a single pre-batched element that
        custom_collate_fn recognizes."""
        data, actual = self.sampler[list(indices)]
        return [(data, actual)]

    def __len__(self):
        return len(self.sampler)

    def get_index(self):
        return self.sampler.get_index()


class DateGroupedBatchSampler(Sampler):
    """One whole trading day per batch (/root/reference/dataset.py:207-238).

    With `rank`/`world_size` set, days are round-robin sharded over DP
    ranks AFTER the (seeded, epoch-synchronized) shuffle — the unit of
    data parallelism.
    """

    def __init__(self, data_source, shuffle: bool = False,
                 rank: int = 0, world_size: int = 1, seed: int = 0):
        self.data_source = data_source
        self.shuffle = shuffle
        self.rank = rank
        self.world_size = world_size
        self.seed = seed
        self.epoch = 0
        self.grouped_indices = self._group_indices_by_date()

    def _group_indices_by_date(self):
        index = self.data_source.sampler.get_index()
        dates = index.get_level_values("datetime")
        codes, _ = pd.factorize(dates, sort=True)
        order = np.arange(len(codes))
        groups = []
        for d in range(codes.max() + 1 if len(codes) else 0):
            groups.append(order[codes == d].tolist())
        return groups

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def _day_order(self):
        n = len(self.grouped_indices)
        order = np.arange(n)
        if self.shuffle:
            rng = np.random.default_rng(self.seed + self.epoch)
            rng.shuffle(order)
        return order

    def _padded_order(self):
        """Day order padded (wrap-around) to a multiple of world_size so
        every rank sees the SAME number of days per epoch: unequal counts
        would make ranks issue different numbers of per-step gradient
        all-reduces (hang / silent corruption at the epoch tail) and
        derive diverging cosine-LR t_max values."""
        order = self._day_order()
        n = len(order)
        if self.world_size > 1 and n % self.world_size != 0 and n > 0:
            pad = self.world_size - n % self.world_size
            reps = (pad + n - 1) // n
            order = np.concatenate([order] + [order] * reps)[:n + pad]
        return order

    def __iter__(self):
        order = self._padded_order()
        for k in range(self.rank, len(order), self.world_size):
            yield self.grouped_indices[order[k]]

    def __len__(self):
        n = len(self.grouped_indices)
        if n == 0:
            return 0
        return (n + self.world_size - 1) // self.world_size


def custom_collate_fn(batch):
    """(data, MultiIndex) pairs -> (tensor (N,T,C+1), list of index lists)
    (/root/reference/dataset.py:242-249). Also accepts the pre-batched
    single element produced by TSDatasetH.__getitems__."""
    if len(batch) == 1 and getattr(batch[0][0], "ndim", 0) == 3:
        data, index = batch[0]
        return torch.as_tensor(data), [list(index)]
    data, indices = zip(*batch)
    data = torch.utils.data.dataloader.default_collate(data)
    indices = [list(index) for index in indices]
    return data, indices


def init_data_loader(df, step_len, shuffle, start, end, select_feature=None,
                     rank: int = 0, world_size: int = 1, seed: int = 0,
                     num_workers: int = 0):
    """Build the day-batched DataLoader (/root/reference/dataset.py:252-274),
    with optional DP sharding.

    pin_memory is intentionally OFF: a day batch is ~4 MB and the epoch
    cache ships each day to HBM exactly once — page-locking every batch
    on a single CPU thread costs far more than the pinned-copy saves
    (measured 10x loader slowdown on the GPU nodes)."""
    if select_feature is not None:
        df = df[select_feature]

    dataset = TSDatasetH(df, step_len=step_len, start=start, end=end,
                         fillna_type="ffill+bfill")
    sampler = DateGroupedBatchSampler(dataset, shuffle=shuffle,
                                      rank=rank, world_size=world_size, seed=seed)
    return DataLoader(
        dataset,
        batch_sampler=sampler,
        collate_fn=custom_collate_fn,
        pin_memory=False,
        num_workers=num_workers,
    )
