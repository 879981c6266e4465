"""Launcher supervision tests: metric aggregation across ranks, failure
detection + restart-from-checkpoint (successor of SURVEY.md §5.3)."""
import os
import time

import pytest

from shifu_amd.config.model_config import ModelConfig
from shifu_amd.config.run_config import RunConfig
from shifu_amd.parallel.launcher import Launcher
from shifu_amd.train.metrics import TrainingIntermediateResult


def _happy_entry(rank, world, rc, mc, sink, heartbeat):
    for epoch in range(2):
        heartbeat()
        sink(TrainingIntermediateResult(worker_index=rank, current_epoch=epoch,
                                        current_epoch_time=0.1 * (rank + 1),
                                        training_error=0.5, valid_error=0.4))


_FAIL_FLAG = None  # set via env in child


def _fail_once_entry(rank, world, rc, mc, sink, heartbeat):
    flag = os.path.join(rc.log_dir, "failed_once")
    if rank == 1 and not os.path.exists(flag):
        open(flag, "w").close()
        raise RuntimeError("injected rank failure")
    for epoch in range(1):
        heartbeat()
        sink(TrainingIntermediateResult(worker_index=rank, current_epoch=epoch,
                                        training_error=0.1, valid_error=0.1,
                                        current_epoch_time=0.05))


def _always_fail_entry(rank, world, rc, mc, sink, heartbeat):
    raise RuntimeError("always fails")


def test_launcher_aggregates_epochs(tmp_path):
    rc = RunConfig(num_gpus=2, log_dir=str(tmp_path))
    la = Launcher(rc, ModelConfig(), _happy_entry)
    stats = la.run()
    assert len(stats) == 2
    assert stats[0].epoch == 0 and stats[1].epoch == 1
    # worker 1 is slower (0.2s vs 0.1s) -> sorted ascending by epoch time
    assert stats[0].workers_by_time == [0, 1]
    board = (tmp_path / "progress.board").read_text()
    assert "epoch 0:" in board and "epoch 1:" in board


def test_launcher_restarts_failed_rank(tmp_path):
    rc = RunConfig(num_gpus=2, log_dir=str(tmp_path), max_rank_restarts=2)
    la = Launcher(rc, ModelConfig(), _fail_once_entry)
    stats = la.run()  # first attempt fails (rank 1), second succeeds
    assert len(stats) >= 1
    assert os.path.exists(tmp_path / "failed_once")


def test_launcher_gives_up_after_max_restarts(tmp_path):
    rc = RunConfig(num_gpus=1, log_dir=str(tmp_path), max_rank_restarts=1)
    la = Launcher(rc, ModelConfig(), _always_fail_entry)
    with pytest.raises(RuntimeError, match="failed after"):
        la.run()


def _slow_start_entry(rank, world, rc, mc, sink, heartbeat):
    time.sleep(3.0)   # longer than the steady-state budget, within startup grace
    heartbeat()
    sink(TrainingIntermediateResult(worker_index=rank, current_epoch=0,
                                    training_error=0.1, valid_error=0.1,
                                    current_epoch_time=0.05))


def _hb_then_hang_entry(rank, world, rc, mc, sink, heartbeat):
    heartbeat()       # ends the startup grace for this rank
    time.sleep(60)


def test_startup_grace_allows_slow_init(tmp_path):
    """Model/arena construction may exceed the heartbeat budget; before a
    rank's first message the (long) startup grace applies (successor of the
    reference's 6-min registration cutover, Constants.java:92-94)."""
    rc = RunConfig(num_gpus=1, log_dir=str(tmp_path), heartbeat_interval_s=0.2,
                   max_missed_heartbeats=5, startup_grace_s=30.0)
    stats = Launcher(rc, ModelConfig(), _slow_start_entry).run()
    assert len(stats) == 1


def test_heartbeat_timeout_after_start(tmp_path):
    """Once a rank has spoken, the normal heartbeat budget applies and a hung
    rank is detected quickly."""
    rc = RunConfig(num_gpus=1, log_dir=str(tmp_path), heartbeat_interval_s=0.2,
                   max_missed_heartbeats=5, startup_grace_s=30.0,
                   max_rank_restarts=0)
    t0 = time.time()
    with pytest.raises(RuntimeError, match="missed heartbeats"):
        Launcher(rc, ModelConfig(), _hb_then_hang_entry).run()
    assert time.time() - t0 < 25.0   # detected via budget, not the 60s sleep
