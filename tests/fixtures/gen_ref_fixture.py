"""Generate the committed miniature reference-schema pickle.

Reproduces the exact output schema of the reference ETL
(/root/reference/data/make_dataset.py:66-83): a pandas DataFrame pickle
with MultiIndex (datetime, instrument), the 158 Alpha158 feature columns
(qlib naming: 9 k-bar + 4 price + 29 rolling ops x 5 windows) and the
CSRankNorm'd LABEL0 column. Values are synthetic (RobustZScore-scale
features, learnable label) — qlib itself is unavailable in the image, so
this fixture pins the schema the loaders must accept, not real market
data. Regenerate with:  python tests/fixtures/gen_ref_fixture.py
"""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)

KBAR = ["KMID", "KLEN", "KMID2", "KUP", "KUP2", "KLOW", "KLOW2", "KSFT",
        "KSFT2"]
PRICE = ["OPEN0", "HIGH0", "LOW0", "VWAP0"]
ROLLING_OPS = ["ROC", "MA", "STD", "BETA", "RSQR", "RESI", "MAX", "MIN",
               "QTLU", "QTLD", "RANK", "RSV", "IMAX", "IMIN", "IMXD",
               "CORR", "CORD", "CNTP", "CNTN", "CNTD", "SUMP", "SUMN",
               "SUMD", "VMA", "VSTD", "WVMA", "VSUMP", "VSUMN", "VSUMD"]
WINDOWS = [5, 10, 20, 30, 60]


def alpha158_columns():
    cols = list(KBAR) + list(PRICE)
    for op in ROLLING_OPS:
        for w in WINDOWS:
            cols.append(f"{op}{w}")
    assert len(cols) == 158, len(cols)
    return cols


def main():
    from factorvae_amd.data.synthetic import make_synthetic_frame

    df = make_synthetic_frame(n_days=50, n_stocks=20, seed=7,
                              start="2019-01-01", label_from_features=True,
                              signal_strength=0.6)
    df.columns = alpha158_columns() + ["LABEL0"]
    out = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "ref_schema_mini.pkl")
    df.to_pickle(out, protocol=4)
    print(f"wrote {out}: {df.shape}, "
          f"{df.index.get_level_values(0).nunique()} days, "
          f"index names {df.index.names}")


if __name__ == "__main__":
    main()
