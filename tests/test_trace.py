"""StepTracer: phase aggregation + chrome trace export + Trainer integration."""
import json
import time

from shifu_amd.utils.trace import StepTracer


def test_tracer_phases_and_export(tmp_path):
    tr = StepTracer(enabled=True, use_gpu_events=False)
    for _ in range(3):
        with tr.phase("fwd"):
            time.sleep(0.002)
        with tr.phase("bwd"):
            time.sleep(0.001)
    st = tr.stats()
    assert st["fwd"]["count"] == 3
    assert st["fwd"]["mean_ms"] >= 1.5
    assert st["bwd"]["mean_ms"] >= 0.5
    assert "fwd=" in tr.summary_line()
    path = str(tmp_path / "trace.json")
    tr.export_chrome_trace(path)
    blob = json.load(open(path))
    assert len(blob["traceEvents"]) == 6
    assert {e["name"] for e in blob["traceEvents"]} == {"fwd", "bwd"}


def test_tracer_disabled_is_noop():
    tr = StepTracer(enabled=False)
    with tr.phase("x"):
        pass
    assert tr.stats() == {}


def test_trainer_trace_integration(tmp_path):
    import torch
    from shifu_amd.config.model_config import ModelConfig
    from shifu_amd.config.run_config import RunConfig
    from shifu_amd.data.csv_loader import TabularDataset
    from shifu_amd.data.synthetic import synthetic_arrays
    from shifu_amd.models.mlp import ShifuMLP
    from shifu_amd.train.trainer import Trainer
    d, c, t, w = synthetic_arrays(400, 6, seed=1)
    full = TabularDataset(d, c, t, w)
    train, valid = full.split(0.2, seed=1)
    mc = ModelConfig.from_dict({"train": {"numTrainEpochs": 1, "params": {
        "NumHiddenLayers": 1, "NumHiddenNodes": [8], "ActivationFunc": ["relu"],
        "LearningRate": 0.01, "Optimizer": "adam", "Loss": "sigmoid_ce",
        "MiniBatchSize": 64, "L2Reg": 0.0}}})
    rc = RunConfig(tmp_model_path=str(tmp_path / "c"),
                   final_model_path=str(tmp_path / "f"),
                   log_dir=str(tmp_path / "logs"), enable_trace=True)
    tr = Trainer(ShifuMLP(6, [8], ["relu"]), mc, rc, train, valid)
    tr.fit()
    st = tr.tracer.stats()
    for phase in ("fwd", "loss", "bwd", "opt"):
        assert phase in st and st[phase]["count"] > 0
    assert (tmp_path / "logs" / "trace-rank0.json").exists()
