"""Observability: metrics logging, step timing, roctx ranges.

The reference's observability is optional wandb logging of per-epoch
losses (/root/reference/main.py:63-66,82-87) plus tqdm bars. This module
provides:

- `MetricsLogger`: wandb when importable AND requested, always a JSONL
  side-file + stdout — same keys the reference logs ("Train Loss",
  "Validation Loss", "Learning Rate", "Best Validation Loss");
- `StepTimer`: device-synchronized rolling timer reporting the headline
  metric (training cross-sections/sec, BASELINE.json);
- `roctx_range`: roctx/nvtx annotation context (rocprofv3 picks these up
  with --marker-trace) for the fused kernel groups — enabled only when
  FACTORVAE_ROCTX=1 since host-side markers are meaningless inside
  hipGraph replay.
"""

from __future__ import annotations

import contextlib
import json
import os
import time
from typing import Optional

import torch

ROCTX_ON = os.environ.get("FACTORVAE_ROCTX", "0") == "1"


@contextlib.contextmanager
def roctx_range(name: str):
    if ROCTX_ON:
        torch.cuda.nvtx.range_push(name)  # roctx on ROCm builds
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class MetricsLogger:
    """Logs per-epoch metrics to stdout + JSONL, optionally wandb."""

    def __init__(self, run_name: str, out_dir: str = ".", use_wandb: bool = False,
                 config: Optional[dict] = None, rank: int = 0):
        self.rank = rank
        self.wandb = None
        self.jsonl = None
        if rank != 0:
            return
        os.makedirs(out_dir, exist_ok=True)
        self.jsonl_path = os.path.join(out_dir, f"{run_name}_metrics.jsonl")
        self.jsonl = open(self.jsonl_path, "a")
        if use_wandb:
            try:
                import wandb  # optional; absent in this image

                wandb.init(project="FactorVAE", name=run_name,
                           config=config or {})
                self.wandb = wandb
            except ImportError:
                pass

    def log(self, metrics: dict, step: Optional[int] = None) -> None:
        if self.rank != 0:
            return
        rec = dict(metrics)
        if step is not None:
            rec["epoch"] = step
        rec["ts"] = time.time()
        self.jsonl.write(json.dumps(rec) + "\n")
        self.jsonl.flush()
        if self.wandb is not None:
            self.wandb.log(metrics, step=step)

    def finish(self, summary: Optional[dict] = None) -> None:
        if self.rank != 0:
            return
        if summary:
            self.log(summary)
        if self.wandb is not None:
            self.wandb.finish()
        if self.jsonl is not None:
            self.jsonl.close()
            self.jsonl = None


class StepTimer:
    """Rolling training-throughput timer (cross-sections/sec)."""

    def __init__(self, device: Optional[torch.device] = None):
        self.device = device
        self.reset()

    def reset(self) -> None:
        self._t0 = None
        self.steps = 0

    def start(self) -> None:
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self._t0 = time.perf_counter()
        self.steps = 0

    def tick(self, n: int = 1) -> None:
        self.steps += n

    def rate(self) -> float:
        """Cross-sections/sec since start() (device-synchronized)."""
        if self._t0 is None or self.steps == 0:
            return 0.0
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        dt = time.perf_counter() - self._t0
        return self.steps / dt if dt > 0 else 0.0
