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


def _ep_crash_once_entry(rank, world, rc, mc, sink, heartbeat):
    """Real training entry: first attempt trains 1 epoch (writing EP shard
    checkpoints) then rank 1 dies; the restart resumes from the shards and
    finishes the full run."""
    import dataclasses
    from shifu_amd.run import default_rank_entry
    marker = os.path.join(rc.log_dir, "crashed_once")
    first = not os.path.exists(marker)
    rc_eff = dataclasses.replace(rc, epochs=1) if first else rc
    default_rank_entry(rank, world, rc_eff, mc, sink, heartbeat)
    if first and rank == 1:
        with open(marker, "w") as f:
            f.write("x")
        raise RuntimeError("injected EP rank failure")


def test_launcher_ep_restart_resumes_from_shards(tmp_path):
    """Whole recovery path end-to-end at world=2 (gloo): EP table-sharded
    unified arenas -> per-rank shard checkpoints -> injected rank death ->
    launcher restart -> resume from the newest complete epoch (shard cache
    skips the CSV re-parse) -> full run completes and exports."""
    from shifu_amd.data.synthetic import generate_synthetic_csv
    from shifu_amd.config.model_config import ModelConfig

    data_dir = tmp_path / "data"
    generate_synthetic_csv(str(data_dir), 600, 4, (23, 31), seed=9, n_files=4)
    mc = ModelConfig.from_dict({
        "train": {"numTrainEpochs": 3, "validSetRate": 0.2,
                  "params": {"NumHiddenLayers": 1, "NumHiddenNodes": [8],
                             "ActivationFunc": ["relu"], "LearningRate": 0.05,
                             "Optimizer": "sgd", "Loss": "sigmoid_ce",
                             "MiniBatchSize": 32, "L2Reg": 0.0}}})
    rc = RunConfig(num_gpus=2, log_dir=str(tmp_path / "logs"),
                   training_data_path=[str(data_dir)],
                   selected_numeric_columns=[2, 3, 4, 5],
                   selected_categorical_columns=[6, 7],
                   vocab_sizes=[23, 31], target_column=0, weight_column=1,
                   tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"),
                   model_type="wide_deep", embed_dim=4, device="cpu",
                   batch_size=32, max_rank_restarts=2,
                   master_port=29763)
    la = Launcher(rc, mc, _ep_crash_once_entry)
    stats = la.run()
    assert len(stats) >= 2, f"expected resumed epochs, got {len(stats)}"
    # per-rank EP shard checkpoints were written
    names = os.listdir(tmp_path / "ckpt")
    assert any("shard0of2" in n for n in names), names
    assert any("shard1of2" in n for n in names), names
    # export happened (chief, after the resumed run)
    assert os.path.exists(tmp_path / "final" / "GenericModelConfig.json")
    # the restart went through the recovery path
    board = (tmp_path / "logs" / "progress.board").read_text()
    assert "restarting from" in board
    # shard cache was populated (the restart's loads skip the re-parse)
    cache = tmp_path / "logs" / "shard_cache"
    assert cache.is_dir() and len(os.listdir(cache)) >= 2
