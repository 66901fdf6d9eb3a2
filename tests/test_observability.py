"""Observability subsystem tests (CPU): JSONL metrics contract,
non-rank-0 no-op, roctx range no-op off-GPU, StepTimer rates.
(SURVEY.md §5.1/§5.5 — the reference has only tqdm/print/wandb.)
"""

import json
import time

from factorvae_amd.observability import MetricsLogger, StepTimer, roctx_range


def test_metrics_logger_jsonl_contract(tmp_path):
    log = MetricsLogger("runx", out_dir=str(tmp_path))
    log.log({"train_loss": 1.5, "val_loss": 1.2}, step=0)
    log.log({"train_loss": 1.1}, step=1)
    log.finish({"best_val": 1.2})
    lines = [json.loads(l) for l in
             open(tmp_path / "runx_metrics.jsonl").read().splitlines()]
    assert len(lines) == 3
    assert lines[0]["train_loss"] == 1.5 and lines[0]["epoch"] == 0
    assert lines[1]["epoch"] == 1
    assert lines[2]["best_val"] == 1.2
    assert all("ts" in l for l in lines)
    # appending across logger instances preserves history (resume)
    log2 = MetricsLogger("runx", out_dir=str(tmp_path))
    log2.log({"train_loss": 0.9}, step=2)
    log2.finish()
    lines = open(tmp_path / "runx_metrics.jsonl").read().splitlines()
    assert len(lines) == 4


def test_metrics_logger_nonzero_rank_is_noop(tmp_path):
    log = MetricsLogger("rk", out_dir=str(tmp_path), rank=1)
    log.log({"x": 1.0}, step=0)  # must not throw, must not write
    log.finish()
    assert not (tmp_path / "rk_metrics.jsonl").exists()


def test_roctx_range_noop_off_gpu():
    with roctx_range("unit"):
        x = 1 + 1
    assert x == 2


def test_step_timer_rate():
    t = StepTimer(device=None)
    assert t.rate() == 0.0  # before start
    t.start()
    time.sleep(0.05)
    t.tick(10)
    r = t.rate()
    assert 0 < r < 10 / 0.05 * 1.5
    t.reset()
    assert t.rate() == 0.0
