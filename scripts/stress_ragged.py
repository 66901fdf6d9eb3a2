"""Ragged-universe stress: many distinct (N, T) day shapes through the
full training driver. This is the reproducer that exposed the ROCm
hipGraphExec-destroy crash fixed in round 2 (see
profiles/r2_final_validation.md and the `_ensure_ws` docstring in
factorvae_amd/engine/fused.py): with per-shape graph eviction, 96
distinct shapes crashed deterministically in epoch 2; with the
synchronized full-reset policy it runs clean.

Run on a GPU box:
  python scripts/stress_ragged.py                  # 96 shapes, cap 256 (no reset)
  FV_WS_CACHE=32 python scripts/stress_ragged.py   # force the overflow-reset path
"""
import argparse
import os
import subprocess
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n_days", type=int, default=160)
    p.add_argument("--n_stocks", type=int, default=400)
    p.add_argument("--n_lo", type=int, default=240,
                   help="min kept stocks per day (wide range => many shapes)")
    p.add_argument("--epochs", type=int, default=5)
    p.add_argument("--seed", type=int, default=7)
    args = p.parse_args()

    import numpy as np
    import pandas as pd

    from factorvae_amd.data.synthetic import make_synthetic_frame

    df = make_synthetic_frame(n_days=args.n_days, n_stocks=args.n_stocks,
                              n_features=158, seed=0)
    rng = np.random.default_rng(args.seed)
    parts = []
    for _, day in df.groupby(level=0):
        keep = rng.integers(args.n_lo, args.n_stocks)
        idx = rng.choice(len(day), size=keep, replace=False)
        parts.append(day.iloc[np.sort(idx)])
    wide = pd.concat(parts)
    shapes = wide.groupby(level=0).size().nunique()
    print(f"{args.n_days} days, {shapes} distinct N in "
          f"[{args.n_lo}, {args.n_stocks})")

    with tempfile.TemporaryDirectory() as tmp:
        data = os.path.join(tmp, "wide.pkl")
        wide.to_pickle(data)
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        r = subprocess.run(
            [sys.executable, os.path.join(repo, "main.py"),
             "--dataset", data, "--num_epochs", str(args.epochs),
             "--run_name", "stress", "--num_factor", "20",
             "--num_portfolio", "128", "--save_dir", tmp,
             "--start_time", "2015-01-01", "--fit_end_time", "2015-06-30",
             "--val_start_time", "2015-07-01", "--val_end_time", "2015-08-15",
             "--end_time", "2015-08-15"],
            env=dict(os.environ, PYTHONFAULTHANDLER="1"))
    sys.exit(r.returncode)


if __name__ == "__main__":
    main()
