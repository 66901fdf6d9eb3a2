"""Backtest / reporting layer (L6 of the reference layer map).

Native re-implementation of what the reference's `backtest.ipynb` gets
from qlib (cells 6-9, /root/reference/backtest.ipynb): a daily top-k
dropout long-only strategy simulator and a `risk_analysis` report
(annualized excess return, information ratio, max drawdown, with and
without transaction costs), plus a driver that goes
checkpoint -> scores -> backtest -> report end-to-end. qlib is not a
dependency: the strategy and the risk statistics follow the published
semantics of qlib's `TopkDropoutStrategy(topk, n_drop)` and
`risk_analysis` so the reported numbers are comparable to BASELINE.md.

Strategy semantics (qlib TopkDropoutStrategy, method_sell="bottom",
method_buy="top", equal-weight):
  every trading day, rank all scoreable stocks by predicted score;
  among current holdings, sell the `n_drop` with the worst ranks; buy
  the best-ranked non-held stocks to refill the portfolio to `topk`
  names; positions are equal-weighted at each rebalance. Costs are
  charged on turnover: `open_cost` on buys, `close_cost` on sells
  (reference run: 5 bp / 15 bp, min-cost ignored at this scale).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import pandas as pd

TRADING_DAYS_PER_YEAR = 238  # qlib's default annualization factor (cn market)


@dataclass
class BacktestConfig:
    topk: int = 50
    n_drop: int = 10
    open_cost: float = 0.0005
    close_cost: float = 0.0015
    risk_free: float = 0.0


@dataclass
class BacktestResult:
    daily_return: pd.Series          # portfolio simple return per day (pre-cost)
    daily_cost: pd.Series            # cost drag per day
    daily_turnover: pd.Series        # fraction of book traded per day
    bench_return: Optional[pd.Series]  # benchmark return per day (if given)
    holdings: List[List[str]] = field(default_factory=list)

    @property
    def excess_no_cost(self) -> pd.Series:
        r = self.daily_return
        if self.bench_return is not None:
            r = r - self.bench_return
        return r.rename("excess_return_without_cost")

    @property
    def excess_with_cost(self) -> pd.Series:
        return (self.excess_no_cost - self.daily_cost).rename(
            "excess_return_with_cost")


def topk_dropout_backtest(
    df: pd.DataFrame,
    score_col: str = "score",
    return_col: str = "LABEL0",
    bench_return: Optional[pd.Series] = None,
    config: Optional[BacktestConfig] = None,
) -> BacktestResult:
    """Simulate the daily top-k dropout strategy over a MultiIndex
    (datetime, instrument) frame holding a score column and a realized
    next-period return column (the reference merges LABEL0 onto the
    scores the same way, backtest.ipynb cell 5).

    Returns per-day portfolio returns, turnover and cost drag.
    """
    cfg = config or BacktestConfig()
    if not isinstance(df.index, pd.MultiIndex):
        raise ValueError("df must be MultiIndex (datetime, instrument)")

    dates = df.index.get_level_values(0).unique().sort_values()
    held: List[str] = []
    rets, costs, turns, hold_log = [], [], [], []

    for date in dates:
        day = df.loc[date]
        scores = day[score_col].dropna()
        ranked = scores.sort_values(ascending=False)
        universe = list(ranked.index)

        if not held:
            target = universe[: cfg.topk]
            n_trade = len(target)  # all buys
            sells: List[str] = []
            buys = list(target)
        else:
            # rank currently-held by today's score; unscored held names
            # rank worst (they cannot be re-validated) and sell first
            held_ranked = sorted(
                held, key=lambda s: (s not in ranked.index,
                                     -(ranked.get(s, -np.inf))))
            n_drop = min(cfg.n_drop, len(held_ranked))
            sells = held_ranked[len(held_ranked) - n_drop:]
            keep = [s for s in held_ranked if s not in set(sells)]
            candidates = [s for s in universe if s not in set(keep)]
            n_buy = min(cfg.topk - len(keep), len(candidates))
            buys = candidates[:n_buy]
            target = keep + buys

        held = target
        hold_log.append(list(held))

        # realized equal-weight return of today's book
        day_rets = day[return_col].reindex(held)
        port_ret = float(day_rets.fillna(0.0).mean()) if held else 0.0

        # turnover & cost: equal-weight book, each traded name is 1/topk
        k = max(len(held), 1)
        buy_frac = len(buys) / k
        sell_frac = len(sells) / k if held else 0.0
        cost = buy_frac * cfg.open_cost + sell_frac * cfg.close_cost
        turn = buy_frac + sell_frac

        rets.append(port_ret)
        costs.append(cost)
        turns.append(turn)

    idx = pd.Index(dates, name="datetime")
    bench = None
    if bench_return is not None:
        bench = bench_return.reindex(idx).fillna(0.0)
    return BacktestResult(
        daily_return=pd.Series(rets, index=idx, name="return"),
        daily_cost=pd.Series(costs, index=idx, name="cost"),
        daily_turnover=pd.Series(turns, index=idx, name="turnover"),
        bench_return=bench,
        holdings=hold_log,
    )


def risk_analysis(r: pd.Series, N: int = TRADING_DAYS_PER_YEAR) -> pd.DataFrame:
    """qlib-compatible risk report of a daily (excess-)return series:
    mean, std, annualized return (mean*N), information ratio
    (mean/std*sqrt(N)), max drawdown of the cumulative-sum curve.
    Matches the statistics printed by backtest.ipynb cell 8."""
    r = r.dropna()
    mean = float(r.mean())
    std = float(r.std(ddof=1))
    annualized_return = mean * N
    information_ratio = mean / std * np.sqrt(N) if std > 0 else np.nan
    cum = r.cumsum()
    max_drawdown = float((cum - cum.cummax()).min()) if len(cum) else np.nan
    return pd.DataFrame(
        {"risk": [mean, std, annualized_return, information_ratio,
                  max_drawdown]},
        index=["mean", "std", "annualized_return", "information_ratio",
               "max_drawdown"],
    )


SIMPLIFICATIONS_NOTE = (
    "simulator: equal-weight top-k dropout with open/close costs only — "
    "unlike the reference's qlib SimulatorExecutor it does NOT model the "
    "9.5% price limit, per-trade minimum cost, or account-level share "
    "rounding; numbers are comparable across models run through THIS "
    "simulator, not directly against qlib-engine outputs"
)


def backtest_report(result: BacktestResult,
                    N: int = TRADING_DAYS_PER_YEAR) -> Dict[str, pd.DataFrame]:
    """The two tables the reference notebook prints (cell 8): risk
    analysis of excess return without and with cost. The 'note' entry
    names this simulator's simplifications vs the reference's qlib
    engine (backtest.ipynb cell 6)."""
    return {
        "excess_return_without_cost": risk_analysis(result.excess_no_cost, N),
        "excess_return_with_cost": risk_analysis(result.excess_with_cost, N),
        "note": SIMPLIFICATIONS_NOTE,
    }


def run_backtest_pipeline(checkpoint: str, data_pickle: str, args,
                          start: Optional[str] = None,
                          end: Optional[str] = None,
                          config: Optional[BacktestConfig] = None):
    """checkpoint -> scores -> merge LABEL0 -> backtest -> report + RankIC.
    Mirror of backtest.ipynb cells 2-9 as one callable."""
    import torch

    from .data.sampler import init_data_loader
    from .utils import RankIC, generate_prediction_scores, load_model

    model = load_model(args)
    state = torch.load(checkpoint, map_location="cpu", weights_only=True)
    model.load_state_dict(state)
    model.eval()

    df = pd.read_pickle(data_pickle)
    loader = init_data_loader(df, step_len=args.seq_length, shuffle=False,
                              start=start, end=end)
    dataset = loader.dataset
    scores = generate_prediction_scores(model, loader, dataset, args)

    merged = scores.join(df[["LABEL0"]], how="inner")
    result = topk_dropout_backtest(merged, config=config)
    report = backtest_report(result)
    rank_ic = RankIC(merged, column1="LABEL0", column2="score")
    return {"scores": scores, "result": result, "report": report,
            "rank_ic": rank_ic}
