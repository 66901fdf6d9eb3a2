"""Reference-layout ETL entrypoint (/root/reference/data/make_dataset.py).

The reference builds its `csi_data.pkl` / `sp500_data.pkl` from qlib's
Alpha158 handler over a local qlib data dump. qlib (and network access
to its data) is not available in this environment, so this script
produces the same OUTPUT CONTRACT from the synthetic generator instead:
a (datetime, instrument) MultiIndex frame with 158 feature columns and
LABEL0, directly loadable by `main.py --dataset <out>`. The exact
reference schema (including qlib's Alpha158 column names) is pinned by
tests/fixtures/ref_schema_mini.pkl and tests/test_ref_fixture.py.

If you DO have qlib and a data dump, run the reference's script to get
real CSI300/S&P500 data — the resulting pickle loads here unchanged.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from factorvae_amd.data.synthetic import make_synthetic_frame

if __name__ == "__main__":
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--out", type=str, default="csi_data.pkl")
    parser.add_argument("--n_days", type=int, default=500)
    parser.add_argument("--n_stocks", type=int, default=300)
    parser.add_argument("--n_features", type=int, default=158)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--start_time", type=str, default="2015-01-01")
    args = parser.parse_args()

    df = make_synthetic_frame(n_days=args.n_days, n_stocks=args.n_stocks,
                              n_features=args.n_features, seed=args.seed,
                              start=args.start_time)
    df.to_pickle(args.out)
    print(f"wrote {args.out}: {df.shape[0]} rows x {df.shape[1]} cols "
          f"({args.n_days} days x {args.n_stocks} stocks, synthetic "
          f"Alpha158-shaped; see docstring re: real qlib data)")
