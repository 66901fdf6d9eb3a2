"""Scoring / backtest CLI — the reference's backtest.ipynb cells 2-9 as
a command-line entry point.

checkpoint -> prediction scores over a date range -> optional CSV dump
(same schema as the reference's released artifacts:
`{run_name}_{K}_{normalize}_{select_feature}_{C}_{H}.csv`, columns
datetime,instrument,score — /root/reference/scores/readme.md) ->
optional top-k dropout backtest + risk report + RankIC.

Run:  python -m factorvae_amd.score --checkpoint best_models/x.pt \
          --dataset data/synthetic.pkl --run_name demo [--backtest]
"""

from __future__ import annotations

import argparse
import os

import pandas as pd
import torch

from .backtest import BacktestConfig, backtest_report, topk_dropout_backtest
from .data.sampler import init_data_loader
from .utils import RankIC, generate_prediction_scores, load_model, test_args


def build_argparser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="Score a test range with a "
                                            "trained FactorVAE checkpoint")
    p.add_argument("--checkpoint", type=str, required=True)
    p.add_argument("--dataset", type=str, required=True, help="data pickle")
    p.add_argument("--run_name", type=str, default="scores")
    p.add_argument("--num_factor", type=int, default=96)
    p.add_argument("--hidden_size", type=int, default=64)
    p.add_argument("--num_latent", type=int, default=158)
    p.add_argument("--num_portfolio", type=int, default=128)
    p.add_argument("--seq_length", type=int, default=20)
    p.add_argument("--start", type=str, default=None)
    p.add_argument("--end", type=str, default=None)
    p.add_argument("--out_dir", type=str, default="./scores")
    p.add_argument("--normalize", action=argparse.BooleanOptionalAction,
                   default=True)
    p.add_argument("--select_feature", action=argparse.BooleanOptionalAction,
                   default=False)
    p.add_argument("--engine", type=str, default="auto",
                   choices=["auto", "fused", "eager"],
                   help="auto = fused HIP predict path on GPU, eager "
                        "module path otherwise")
    p.add_argument("--backtest", action="store_true",
                   help="run the top-k dropout backtest + risk report")
    p.add_argument("--report", action="store_true",
                   help="with --backtest: also write backtest.png and the "
                        "plotly HTML report (the reference notebook's "
                        "backtest.png / backtest_plotly/ artifacts)")
    p.add_argument("--topk", type=int, default=50)
    p.add_argument("--n_drop", type=int, default=10)
    return p


def main(argv=None):
    args = build_argparser().parse_args(argv)
    targs = test_args(
        run_name=args.run_name, num_factor=args.num_factor,
        hidden_size=args.hidden_size, num_latent=args.num_latent,
        num_portfolio=args.num_portfolio, seq_length=args.seq_length,
    )
    model = load_model(targs)
    state = torch.load(args.checkpoint, map_location="cpu", weights_only=True)
    model.load_state_dict(state)
    model.eval()

    df = pd.read_pickle(args.dataset)
    loader = init_data_loader(df, step_len=args.seq_length, shuffle=False,
                              start=args.start, end=args.end)
    scores = generate_prediction_scores(model, loader, loader.dataset,
                                        targs, engine=args.engine)

    os.makedirs(args.out_dir, exist_ok=True)
    # reference artifact name schema (scores/readme.md)
    fname = (f"{args.run_name}_{args.num_factor}_{args.normalize}_"
             f"{args.select_feature}_{args.num_latent}_{args.hidden_size}.csv")
    out_csv = os.path.join(args.out_dir, fname)
    scores.to_csv(out_csv)
    print(f"wrote {out_csv}: {len(scores)} rows")

    if args.backtest:
        merged = scores.join(df[["LABEL0"]], how="inner")
        result = topk_dropout_backtest(
            merged, config=BacktestConfig(topk=args.topk, n_drop=args.n_drop))
        report = backtest_report(result)
        print("\nexcess_return_without_cost:")
        print(report["excess_return_without_cost"])
        print("\nexcess_return_with_cost:")
        print(report["excess_return_with_cost"])
        print(f"\nNOTE: {report['note']}")
        print("\nRankIC:")
        print(RankIC(merged, column1="LABEL0", column2="score"))
        if args.report:
            from .report import write_backtest_png, write_plotly_report

            png = write_backtest_png(
                result, os.path.join(args.out_dir, "backtest.png"),
                title=f"{args.run_name} backtest")
            html = write_plotly_report(
                result, os.path.join(args.out_dir, "backtest_plotly"))
            print(f"wrote {png} and {html}")
    return scores


if __name__ == "__main__":
    main()
