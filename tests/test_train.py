"""Training-loop, determinism, and scoring tests (CPU)."""

import numpy as np
import pandas as pd
import torch

from factorvae_amd.data.sampler import init_data_loader
from factorvae_amd.data.synthetic import make_synthetic_frame
from factorvae_amd.engine.trainer import train, train_main, validate
from factorvae_amd.models.modules import build_factorvae
from factorvae_amd.utils import DataArgument, RankIC, generate_prediction_scores, set_seed


class Args:
    num_epochs = 2
    lr = 1e-3
    num_latent = 10
    num_portfolio = 12
    seq_len = 5
    num_factor = 4
    hidden_size = 8
    seed = 42
    run_name = "unit"
    num_workers = 0
    wandb = False


def _frame():
    return make_synthetic_frame(n_days=40, n_stocks=12, n_features=10, seed=0,
                                start="2015-01-01")


def test_train_epoch_runs_and_loss_finite():
    set_seed(0)
    df = _frame()
    dates = df.index.levels[0]
    loader = init_data_loader(df, step_len=5, shuffle=True, start=dates[0], end=dates[29])
    model = build_factorvae(num_latent=10, hidden_size=8, num_portfolio=12, num_factor=4)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    sched = torch.optim.lr_scheduler.CosineAnnealingLR(opt, T_max=len(loader) * 2)
    device = torch.device("cpu")
    l0 = train(model, loader, opt, sched, device=device)
    l1 = train(model, loader, opt, sched, device=device)
    assert np.isfinite(l0) and np.isfinite(l1)
    v = validate(model, loader, device=device)
    assert np.isfinite(v)


def test_set_seed_determinism():
    df = _frame()
    dates = df.index.levels[0]

    def run():
        set_seed(7)
        loader = init_data_loader(df, step_len=5, shuffle=True, start=dates[0],
                                  end=dates[29], seed=7)
        model = build_factorvae(num_latent=10, hidden_size=8, num_portfolio=12, num_factor=4)
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        return train(model, loader, opt, None, device=torch.device("cpu"))

    assert run() == run()


def test_train_main_saves_best_checkpoint(tmp_path):
    df = _frame()
    args = Args()
    args.save_dir = str(tmp_path)
    args.dataset = None
    data_args = DataArgument(start_time="2015-01-01", fit_end_time="2015-02-10",
                             val_start_time="2015-02-11", val_end_time="2015-02-25",
                             end_time="2015-02-25", seq_len=5)
    best = train_main(args, data_args, df=df)
    assert np.isfinite(best)
    ckpts = list(tmp_path.glob("*.pt"))
    assert len(ckpts) == 1
    assert ckpts[0].name == "unit_factor_4_hdn_8_port_12_seed_42.pt"
    sd = torch.load(ckpts[0], weights_only=True)
    model = build_factorvae(num_latent=10, hidden_size=8, num_portfolio=12, num_factor=4)
    model.load_state_dict(sd)  # loads cleanly


def test_generate_prediction_scores_and_rankic():
    set_seed(1)
    df = _frame()
    dates = df.index.levels[0]
    loader = init_data_loader(df, step_len=5, shuffle=False, start=dates[10], end=dates[-1])
    model = build_factorvae(num_latent=10, hidden_size=8, num_portfolio=12, num_factor=4)

    class A:
        seq_length = 5

    test_ds = loader.dataset
    scores = generate_prediction_scores(model, loader, test_ds, A())
    assert list(scores.columns) == ["score"]
    assert scores.index.names == ["datetime", "instrument"]
    assert len(scores) == len(test_ds.get_index())

    merged = scores.join(df["LABEL0"])
    out = RankIC(merged, column1="LABEL0", column2="score")
    assert "RankIC" in out.columns and "RankIC_IR" in out.columns
    assert np.isfinite(out["RankIC"].iloc[0])


def test_rankic_perfect_correlation():
    idx = pd.MultiIndex.from_product(
        [pd.bdate_range("2020-01-01", periods=3), [f"s{i}" for i in range(10)]],
        names=["datetime", "instrument"],
    )
    vals = np.tile(np.arange(10, dtype=float), 3)
    df = pd.DataFrame({"LABEL0": vals, "Pred": vals * 2 + 1}, index=idx)
    out = RankIC(df)
    assert abs(out["RankIC"].iloc[0] - 1.0) < 1e-9


def test_train_h128_eager_fallback_target():
    """H=128 exceeds the fused kernels' envelope (VERDICT weak #3): the
    resolver routes to eager — prove the eager target actually trains at
    H=128 (finite losses, one full epoch) so the fallback is a working
    path, not just a different error."""
    set_seed(3)
    df = _frame()
    dates = df.index.levels[0]
    loader = init_data_loader(df, step_len=5, shuffle=True, start=dates[0],
                              end=dates[19])
    model = build_factorvae(num_latent=10, hidden_size=128, num_portfolio=12,
                            num_factor=4)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    device = torch.device("cpu")
    loss = train(model, loader, opt, None, device=device)
    assert np.isfinite(loss)
    assert np.isfinite(validate(model, loader, device=device))


def test_root_entrypoint_shims(tmp_path):
    """Reference-layout drop-ins: `python main.py` and
    `from train_model import train` work from the repo root
    (/root/reference/main.py, train_model.py)."""
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    import importlib.util as iu

    spec = iu.spec_from_file_location("train_model",
                                     os.path.join(root, "train_model.py"))
    tm = iu.module_from_spec(spec)
    spec.loader.exec_module(tm)
    assert callable(tm.train) and callable(tm.validate) and callable(tm.test)

    df = _frame()
    data = tmp_path / "d.pkl"
    df.to_pickle(data)
    r = subprocess.run(
        [sys.executable, "main.py", "--num_epochs", "1", "--run_name", "sh",
         "--dataset", str(data), "--num_latent", "10", "--hidden_size", "8",
         "--num_portfolio", "12", "--num_factor", "4", "--seq_len", "5",
         "--num_workers", "0", "--save_dir", str(tmp_path),
         "--start_time", "2015-01-01", "--fit_end_time", "2015-02-10",
         "--val_start_time", "2015-02-11", "--end_time", "2015-03-01",
         "--val_end_time", "2015-03-01"],
        cwd=root, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]


def test_root_module_shims_reference_imports():
    """The reference's import lines work verbatim against this repo:
    `from module import FactorVAE`, `from dataset import
    init_data_loader`, `from utils import RankIC` — every public name
    of /root/reference/{module,dataset,utils}.py resolves (except the
    dead FactorVAE_old, dropped per SURVEY §2.1 #16)."""
    import importlib
    import os
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, root)
    try:
        surface = {
            "module": ["FeatureExtractor", "FactorEncoder", "AlphaLayer",
                       "BetaLayer", "FactorDecoder", "AttentionLayer",
                       "FactorPredictor", "FactorVAE"],
            "dataset": ["np_ffill", "TSDataSampler", "TSDatasetH",
                        "DateGroupedBatchSampler", "custom_collate_fn",
                        "init_data_loader"],
            "utils": ["set_seed", "DataArgument", "load_model",
                      "generate_prediction_scores", "test_args", "RankIC"],
        }
        for m, names in surface.items():
            mod = importlib.import_module(m)
            for n in names:
                assert hasattr(mod, n), (m, n)
        from module import FactorVAE

        from factorvae_amd.models.modules import FactorVAE as Impl
        assert FactorVAE is Impl
    finally:
        sys.path.remove(root)
