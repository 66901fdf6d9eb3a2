"""Utilities: seeding, config dataclasses, model loading, scoring, RankIC.

Functional equivalents of /root/reference/utils.py:10-129 (new code,
same contracts).
"""

from __future__ import annotations

import os
import random
from dataclasses import dataclass, field
from typing import Optional

import numpy as np
import pandas as pd
import torch
from scipy.stats import spearmanr


def set_seed(seed: int) -> None:
    """Determinism contract of the reference (/root/reference/utils.py:10-17)."""
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
        torch.cuda.manual_seed_all(seed)
    torch.backends.cudnn.deterministic = True
    torch.backends.cudnn.benchmark = False


@dataclass
class DataArgument:
    """Data-range config (/root/reference/utils.py:19-53)."""

    save_dir: str = "./data"
    start_time: str = "2010-12-01"
    end_time: str = "2020-12-31"
    fit_end_time: str = "2017-12-31"
    val_start_time: str = "2018-01-01"
    val_end_time: str = "2018-12-31"
    seq_len: int = 20
    normalize: bool = True
    select_feature: Optional[str] = None


@dataclass
class test_args:
    """Inference config (/root/reference/utils.py:95-110)."""

    run_name: str
    num_factor: int
    normalize: bool = True
    select_feature: bool = True

    batch_size: int = 300
    seq_length: int = 20

    hidden_size: int = 20
    num_latent: int = 20
    num_portfolio: int = 128

    save_dir: str = "./best_model"
    use_qlib: bool = False


def load_model(args):
    """Rebuild the 6-module FactorVAE from an args object
    (/root/reference/utils.py:57-67)."""
    from .models.modules import build_factorvae

    return build_factorvae(
        num_latent=args.num_latent,
        hidden_size=args.hidden_size,
        num_portfolio=args.num_portfolio,
        num_factor=args.num_factor,
    )


def checkpoint_path(save_dir: str, run_name: str, num_factor: int,
                    hidden_size: int, num_portfolio: int, seed: int) -> str:
    """Checkpoint file-name schema of the reference (/root/reference/main.py:78)."""
    return os.path.join(
        save_dir,
        f"{run_name}_factor_{num_factor}_hdn_{hidden_size}_port_{num_portfolio}_seed_{seed}.pt",
    )


@torch.no_grad()
def generate_prediction_scores(model, test_dataloader, test_dataset, args,
                               engine: str = "auto"):
    """Score every test cross-section with model.prediction; returns a
    MultiIndex (datetime, instrument) DataFrame['score'].

    Matches the operative scorer of the reference (backtest.ipynb cell 1,
    which slices the label column off the (N,T,C+1) block; the
    utils.py:70-93 variant is shadowed by it).

    engine="auto" uses the fused HIP predict path on GPU (extractor ->
    prior predictor -> stochastic decoder, same math incl. the eval-time
    reparameterized sample); "eager" forces the module path."""
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    model.to(device)
    model.eval()
    ls = []

    trainer = None
    if engine == "auto" and device.type == "cuda":
        try:
            from .engine.fused import FusedTrainer

            trainer = FusedTrainer(model, lr=0.0, t_max=1, device=device,
                                   train=False)
        except Exception:
            trainer = None

    for char_with_label, _ in test_dataloader:
        if char_with_label.shape[1] != args.seq_length:
            continue
        char = char_with_label[:, :, :-1].to(device)
        if trainer is not None:
            predictions = trainer.predict(char.float())
        else:
            predictions = model.prediction(char.float())
        ls.append(predictions.detach().cpu())

    ls = torch.cat(ls, dim=0)
    multi_index = pd.MultiIndex.from_tuples(
        test_dataset.get_index(), names=["datetime", "instrument"]
    )
    return pd.DataFrame(ls.numpy(), index=multi_index, columns=["score"])


def RankIC(df: pd.DataFrame, column1: str = "LABEL0", column2: str = "Pred"):
    """Per-day Spearman rank IC; returns DataFrame with RankIC and
    RankIC_IR = mean/std (/root/reference/utils.py:113-129)."""
    ric_values = []
    for date in df.index.get_level_values(0).unique():
        daily = df.loc[date]
        ric, _ = spearmanr(daily[column1].rank(), daily[column2].rank())
        ric_values.append(ric)

    if not ric_values:
        # same type as the populated case (callers index/print a frame)
        return pd.DataFrame({"RankIC": [np.nan], "RankIC_IR": [np.nan]})

    ric = np.mean(ric_values)
    std = np.std(ric_values)
    ir = ric / std if std != 0 else np.nan
    return pd.DataFrame({"RankIC": [ric], "RankIC_IR": [ir]})
