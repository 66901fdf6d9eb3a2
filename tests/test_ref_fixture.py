"""End-to-end test over the committed reference-schema pickle fixture.

The fixture (tests/fixtures/ref_schema_mini.pkl) pins the reference
ETL's exact output schema (/root/reference/data/make_dataset.py:66-83):
MultiIndex (datetime, instrument) DataFrame, 158 Alpha158-named feature
columns + LABEL0. This test exercises the FILE path the CLI uses
(pd.read_pickle via args.dataset) through train -> checkpoint ->
score CSV -> backtest report, i.e. the full reference workflow
(main.py -> backtest.ipynb) on our engine.
"""

import os
from types import SimpleNamespace

import pandas as pd
import pytest

FIX = os.path.join(os.path.dirname(os.path.abspath(__file__)), "fixtures",
                   "ref_schema_mini.pkl")


def test_fixture_matches_reference_schema():
    df = pd.read_pickle(FIX)
    assert list(df.index.names) == ["datetime", "instrument"]
    assert df.shape[1] == 159
    assert df.columns[-1] == "LABEL0"
    # qlib Alpha158 naming: k-bar + price + rolling-op families
    assert list(df.columns[:4]) == ["KMID", "KLEN", "KMID2", "KUP"]
    assert "ROC5" in df.columns and "VSUMD60" in df.columns
    assert df.index.get_level_values(0).nunique() == 50
    # the loader contract main.py applies: first 159 cols, last renamed
    loaded = pd.read_pickle(FIX).iloc[:, :159]
    loaded = loaded.rename(columns={loaded.columns[-1]: "LABEL0"})
    assert loaded.equals(df)


@pytest.mark.timeout(600)
def test_train_score_backtest_over_fixture(tmp_path):
    from factorvae_amd import score as score_cli
    from factorvae_amd.engine.trainer import train_main
    from factorvae_amd.utils import DataArgument, checkpoint_path

    args = SimpleNamespace(
        num_epochs=2, lr=2e-3, num_latent=158, num_portfolio=8,
        seq_len=5, num_factor=4, hidden_size=16, seed=0,
        run_name="fixture_e2e", save_dir=str(tmp_path), dataset=str(FIX),
        engine="eager", wandb=False, resume=False,
    )
    data_args = DataArgument(
        start_time="2019-01-01", end_time="2019-12-31",
        fit_end_time="2019-02-20", val_start_time="2019-02-21",
        val_end_time="2019-03-12", seq_len=5,
    )
    best = train_main(args, data_args, df=None)  # df=None -> file path
    assert best == best  # finite
    ckpt = checkpoint_path(str(tmp_path), "fixture_e2e", 4, 16, 8, 0)
    assert os.path.exists(ckpt)

    out_dir = str(tmp_path / "scores")
    scores = score_cli.main([
        "--checkpoint", ckpt, "--dataset", FIX, "--run_name", "fixture",
        "--num_factor", "4", "--hidden_size", "16", "--num_latent", "158",
        "--num_portfolio", "8", "--seq_length", "5",
        "--out_dir", out_dir, "--backtest", "--topk", "5", "--n_drop", "2",
    ])
    # reference artifact name schema {run}_{K}_{norm}_{feat}_{C}_{H}.csv
    csv = os.path.join(out_dir, "fixture_4_True_False_158_16.csv")
    assert os.path.exists(csv)
    out = pd.read_csv(csv)
    assert list(out.columns) == ["datetime", "instrument", "score"]
    assert len(out) == len(scores) > 0
