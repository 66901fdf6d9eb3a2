"""CLI entry point — same flag surface as /root/reference/main.py:89-114.

Run:  python -m factorvae_amd.main [flags]
Distributed (8x MI355X, one rank per GPU over RCCL):
      python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
          --master-addr 127.0.0.1 -m factorvae_amd.main [flags]
"""

from __future__ import annotations

import argparse

from .engine.trainer import train_main
from .utils import DataArgument


def build_argparser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(description="Train a FactorVAE model on stock data")

    parser.add_argument("--num_epochs", type=int, default=30, help="number of epochs to train for")
    parser.add_argument("--lr", type=float, default=0.0001, help="learning rate")

    parser.add_argument("--num_latent", type=int, default=158, help="number of input features C")
    parser.add_argument("--num_portfolio", type=int, default=128, help="number of portfolios M")

    parser.add_argument("--seq_len", type=int, default=20, help="sequence length T")
    parser.add_argument("--num_factor", type=int, default=96, help="number of factors K")
    parser.add_argument("--hidden_size", type=int, default=64, help="hidden size H")

    parser.add_argument("--dataset", type=str, default="./data/csi_data.pkl", help="dataset pickle")
    parser.add_argument("--start_time", type=str, default="2009-01-01")
    parser.add_argument("--fit_end_time", type=str, default="2017-12-31")
    parser.add_argument("--val_start_time", type=str, default="2018-01-01")
    parser.add_argument("--val_end_time", type=str, default="2018-12-31")
    parser.add_argument("--end_time", type=str, default="2020-12-31")

    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--run_name", type=str, default="VAE-Revision2")
    parser.add_argument("--save_dir", type=str, default="./best_models")
    parser.add_argument("--num_workers", type=int, default=4)
    parser.add_argument("--wandb", action="store_true",
                        help="log to wandb if installed (no-op otherwise)")
    parser.add_argument("--engine", type=str, default="auto",
                        choices=["auto", "fused", "eager"],
                        help="auto = fused HIP engine on GPU, eager on CPU")
    parser.add_argument("--dtype", type=str, default="fp32",
                        choices=["fp32", "bf16", "fp8"],
                        help="fused-engine compute dtype for the extractor "
                             "GEMM family (bf16 MFMA path)")
    parser.add_argument("--resume", action="store_true",
                        help="resume from the checkpoint + optimizer "
                             "side-car in save_dir if present")
    return parser


def main(argv=None) -> float:
    args = build_argparser().parse_args(argv)
    data_args = DataArgument(
        start_time=args.start_time,
        end_time=args.end_time,
        fit_end_time=args.fit_end_time,
        val_start_time=args.val_start_time,
        val_end_time=args.val_end_time,
        seq_len=args.seq_len,
    )
    return train_main(args, data_args)


if __name__ == "__main__":
    main()
