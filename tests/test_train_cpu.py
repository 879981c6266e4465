"""End-to-end CPU world_size=1 training (BASELINE.json config 1):
3-layer MLP binary classifier on synthetic CSV through the full
ModelConfig/ColumnConfig plumbing, plus checkpoint/resume, export/scorer,
Wide&Deep / DeepFM smoke, and metric aggregation."""
import os

import numpy as np
import pytest
import torch

from shifu_amd.config.model_config import ModelConfig
from shifu_amd.config.run_config import RunConfig
from shifu_amd.data.csv_loader import TabularDataset, load_csv_files
from shifu_amd.data.synthetic import generate_synthetic_csv, synthetic_arrays
from shifu_amd.models.deepfm import DeepFM
from shifu_amd.models.mlp import ShifuMLP
from shifu_amd.models.wide_deep import WideDeep
from shifu_amd.train.checkpoint import latest_checkpoint
from shifu_amd.train.export import load_exported
from shifu_amd.train.metrics import EpochStats, TrainingIntermediateResult
from shifu_amd.train.trainer import Trainer, auc_score
from shifu_amd.serve import ShifuScorer


def _mc(epochs=3, loss="sigmoid_ce", opt="adam", lr=0.01, hidden=(16, 8),
        acts=("relu", "relu"), batch=64):
    return ModelConfig.from_dict({
        "train": {"numTrainEpochs": epochs, "validSetRate": 0.2,
                  "params": {"NumHiddenLayers": len(hidden),
                             "NumHiddenNodes": list(hidden),
                             "ActivationFunc": list(acts),
                             "LearningRate": lr, "Optimizer": opt,
                             "Loss": loss, "MiniBatchSize": batch,
                             "L2Reg": 0.0}}})


def _data(n=1200, n_dense=10, vocab=(), seed=5):
    dense, cats, target, weight = synthetic_arrays(n, n_dense, vocab, seed=seed)
    full = TabularDataset(dense, cats, target, weight)
    return full.split(0.2, seed=1)


def test_mlp_end_to_end_learns(tmp_path):
    train, valid = _data()
    mc = _mc()
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    model = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    tr = Trainer(model, mc, rc, train, valid)
    first = tr.evaluate(tr.valid_data)
    results = tr.fit()
    last = tr.evaluate(tr.valid_data)
    assert len(results) == 3
    assert last["loss"] < first["loss"]
    assert last["auc"] > 0.6  # learns the synthetic signal
    # checkpoint + export artifacts exist
    assert latest_checkpoint(str(tmp_path / "ckpt")) is not None
    assert os.path.exists(tmp_path / "final" / "GenericModelConfig.json")


def test_resume_from_checkpoint(tmp_path):
    train, valid = _data()
    mc = _mc(epochs=2)
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    model = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    tr = Trainer(model, mc, rc, train, valid)
    tr.fit()
    step_after_2 = tr.global_step

    # a fresh trainer resumes at epoch 2 and runs only the remaining epochs
    mc2 = _mc(epochs=4)
    model2 = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=99)
    tr2 = Trainer(model2, mc2, rc, train, valid)
    results = tr2.fit()
    assert tr2.start_epoch == 2
    assert [r.current_epoch for r in results] == [2, 3]
    assert tr2.global_step > step_after_2


def test_export_scorer_roundtrip(tmp_path):
    train, valid = _data(n=600)
    mc = _mc(epochs=1)
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    model = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    tr = Trainer(model, mc, rc, train, valid)
    tr.fit()

    # reload via the export layout and check score parity
    m2 = load_exported(str(tmp_path / "final"))
    x = torch.from_numpy(valid.dense[:32])
    p1 = model.predict(x).numpy()
    p2 = m2.predict(x).numpy()
    assert np.allclose(p1, p2, atol=1e-6)

    # single-row scorer (the Computable-equivalent path,
    # TensorflowModel.java:52-94)
    sc = ShifuScorer()
    sc.init(str(tmp_path / "final" / "GenericModelConfig.json"))
    p3 = sc.compute(valid.dense[0].tolist())
    assert 0.0 <= p3 <= 1.0
    assert abs(p3 - float(p1[0])) < 1e-5


@pytest.mark.parametrize("cls", [WideDeep, DeepFM])
def test_embedding_models_learn(tmp_path, cls):
    train, valid = _data(n=1500, n_dense=6, vocab=(30, 50), seed=11)
    mc = _mc(epochs=4, lr=0.02)
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    model = cls(6, [30, 50], 8, [16, 8], ["relu", "relu"], seed=2)
    tr = Trainer(model, mc, rc, train, valid)
    first = tr.evaluate(tr.valid_data)
    tr.fit()
    last = tr.evaluate(tr.valid_data)
    assert last["loss"] < first["loss"]
    assert os.path.exists(tmp_path / "final" / "graph.json")
    m2 = load_exported(str(tmp_path / "final"))
    d = torch.from_numpy(valid.dense[:8])
    c = torch.from_numpy(valid.cats[:8])
    assert np.allclose(model.predict(d, c).numpy(), m2.predict(d, c).numpy(), atol=1e-6)


def test_full_csv_pipeline(tmp_path):
    """CSV on disk -> loader -> trainer (config 1 wiring)."""
    paths = generate_synthetic_csv(str(tmp_path / "data"), n_rows=800, n_dense=5,
                                   seed=21)
    ds = load_csv_files(paths, selected_numeric=[2, 3, 4, 5, 6],
                        target_column=0, weight_column=1)
    train, valid = ds.split(0.2, seed=1)
    mc = _mc(epochs=2)
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    tr = Trainer(ShifuMLP(5, [16, 8], ["relu", "relu"]), mc, rc, train, valid)
    results = tr.fit()
    assert all(r.training_error == r.training_error for r in results)  # no NaN


def test_window_mode_runs(tmp_path):
    """SAGN-style local window (update_window=5) trains without sync every step."""
    train, valid = _data(n=800)
    mc = _mc(epochs=2)
    mc.params.update_window = 5
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    tr = Trainer(ShifuMLP(10, [16, 8], ["relu", "relu"]), mc, rc, train, valid)
    first = tr.evaluate(tr.valid_data)
    tr.fit()
    assert tr.evaluate(tr.valid_data)["loss"] < first["loss"]


def test_metric_line_roundtrip():
    r = TrainingIntermediateResult(worker_index=3, current_epoch=7,
                                   current_epoch_time=1.5, current_epoch_valid_time=0.25,
                                   training_error=0.125, valid_error=0.25)
    r2 = TrainingIntermediateResult.from_line(r.to_line())
    assert r2 == TrainingIntermediateResult(3, 7, 1.5, 0.25, 0.125, 0.25, "")


def test_epoch_stats_aggregation():
    rs = [TrainingIntermediateResult(i, 1, float(10 - i), 0.5, 0.1 * i, 0.2 * i)
          for i in range(4)]
    st = EpochStats.aggregate(rs)
    assert st.epoch == 1
    assert abs(st.mean_training_error - np.mean([0.0, 0.1, 0.2, 0.3])) < 1e-9
    assert st.workers_by_time == [3, 2, 1, 0]  # sorted by epoch time (doStatistic)


def test_auc_score_sanity():
    labels = np.array([0, 0, 1, 1], dtype=np.float32)
    assert auc_score(np.array([0.1, 0.2, 0.8, 0.9]), labels) == 1.0
    assert auc_score(np.array([0.9, 0.8, 0.2, 0.1]), labels) == 0.0
    assert abs(auc_score(np.array([0.5, 0.5, 0.5, 0.5]), labels) - 0.5) < 1e-9


def test_per_epoch_aggregation_semantics(tmp_path):
    """The reference's headline mode applies ONE global update per epoch
    (SyncReplicasOptimizer with replicas_to_aggregate = total_rows/batch,
    ssgd_monitor.py:139-140).  update_window >= steps/epoch reproduces it:
    exactly one optimizer step per epoch."""
    train, valid = _data(n=640)
    mc = _mc(epochs=3, batch=64)
    mc.params.update_window = 10 ** 6
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    tr = Trainer(ShifuMLP(10, [16, 8], ["relu", "relu"]), mc, rc, train, valid)
    tr.fit()
    assert tr.optimizer.step_count == 3  # one aggregated update per epoch


def test_reference_default_combo(tmp_path):
    """The reference's production defaults: weighted-MSE loss on a sigmoid
    head + Adadelta(lr=1.0) + L2(0.1) (ssgd_monitor.py:129,138,58-68)."""
    train, valid = _data(n=1500)
    mc = _mc(epochs=6, loss="weighted_mse", opt="adadelta", lr=1.0)
    mc.params.l2_reg = 0.1
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    tr = Trainer(ShifuMLP(10, [30, 20], ["tanh", "tanh"], seed=5), mc, rc,
                 train, valid)
    first = tr.evaluate(tr.valid_data)
    tr.fit()
    last = tr.evaluate(tr.valid_data)
    assert last["loss"] < first["loss"]
    assert last["auc"] > 0.55


def test_time_based_checkpointing(tmp_path):
    """rc.checkpoint_every_secs triggers mid-epoch saves (Supervisor
    save_model_secs successor, ssgd.py:124-128)."""
    train, valid = _data(n=3000)
    mc = _mc(epochs=2, batch=16)      # many small steps
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"),
                   checkpoint_every_secs=0.05)
    tr = Trainer(ShifuMLP(10, [16, 8], ["relu", "relu"]), mc, rc, train, valid)
    tr.fit()
    assert latest_checkpoint(str(tmp_path / "ckpt")) is not None


def test_resume_rng_fast_forward(tmp_path):
    """A checkpoint without a saved numpy RNG stream (non-chief ranks; legacy
    checkpoints) must fast-forward the permutation stream so the resumed run
    draws the same epoch orderings an uninterrupted run would."""
    train, valid = _data(n=400)
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"))
    model = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    tr = Trainer(model, _mc(epochs=2), rc, train, valid)
    tr.fit()

    path = latest_checkpoint(str(tmp_path / "ckpt"))
    blob = torch.load(path, weights_only=False)
    blob["extra"].pop("np_rng")
    torch.save(blob, path)

    model2 = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    tr2 = Trainer(model2, _mc(epochs=4), rc, train, valid)
    tr2.maybe_resume()
    assert tr2.start_epoch == 2

    ref = np.random.default_rng(rc.seed + 0)
    n = len(tr2.train_data)
    for _ in range(2):
        ref.permutation(n)
    assert np.array_equal(tr2._rng.permutation(n), ref.permutation(n))


def test_resume_equivalence_cpu(tmp_path):
    """4 epochs straight vs 2 + resume + 2 on CPU fp32: identical final
    valid loss (exact determinism — no atomics on the CPU path)."""
    train, valid = _data(n=400)

    def run(epochs, sub):
        rc = RunConfig(tmp_model_path=str(tmp_path / sub / "ckpt"),
                       final_model_path=str(tmp_path / sub / "final"))
        model = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
        tr = Trainer(model, _mc(epochs=epochs), rc, train, valid)
        tr.fit()
        return tr

    straight = run(4, "a").evaluate(run(4, "a2").valid_data)["loss"]

    rc_b = RunConfig(tmp_model_path=str(tmp_path / "b" / "ckpt"),
                     final_model_path=str(tmp_path / "b" / "final"))
    m1 = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    Trainer(m1, _mc(epochs=2), rc_b, train, valid).fit()
    m2 = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
    tr_res = Trainer(m2, _mc(epochs=4), rc_b, train, valid)
    tr_res.fit()
    resumed = tr_res.evaluate(tr_res.valid_data)["loss"]
    assert abs(straight - resumed) < 1e-7, (straight, resumed)


def test_streaming_residency_matches_device(tmp_path):
    """data_residency="stream" (host-resident shard, per-batch transfer) must
    train identically to the default device-resident path."""
    train, valid = _data(n=400)
    mc = _mc(epochs=2)

    def run(residency, sub):
        rc = RunConfig(tmp_model_path=str(tmp_path / sub / "ckpt"),
                       final_model_path=str(tmp_path / sub / "final"),
                       data_residency=residency)
        model = ShifuMLP(10, [16, 8], ["relu", "relu"], seed=3)
        tr = Trainer(model, mc, rc, train, valid)
        tr.fit()
        return tr.evaluate(tr.valid_data)["loss"]

    resident = run("auto", "a")
    streamed = run("stream", "b")
    assert abs(resident - streamed) < 1e-7, (resident, streamed)


def test_run_to_run_determinism(tmp_path):
    """Two identical CPU runs (same seeds/config, fresh dirs) must produce
    BITWISE identical final parameters — the determinism guarantee the
    precision model documents (ARCHITECTURE §8) at world=1."""
    def run(tag):
        train, valid = _data(800, 6, (20, 30), seed=7)
        rc = RunConfig(tmp_model_path=str(tmp_path / f"ck{tag}"),
                       final_model_path=str(tmp_path / f"fn{tag}"),
                       log_dir=str(tmp_path / f"lg{tag}"), device="cpu",
                       model_type="wide_deep", embed_dim=4, seed=3,
                       vocab_sizes=[20, 30])
        from shifu_amd.models.mlp import build_model
        model = build_model(_mc(epochs=2), 6, [20, 30],
                            model_type="wide_deep", embed_dim=4, seed=3)
        tr = Trainer(model, _mc(epochs=2), rc, train, valid)
        tr.fit()
        return {k: v.detach().clone() for k, v in model.state_dict().items()}

    s1 = run("a")
    s2 = run("b")
    assert s1.keys() == s2.keys()
    for k in s1:
        assert torch.equal(s1[k], s2[k]), f"nondeterministic param {k}"


def test_auc_matches_sklearn():
    """auc_score (rank-based, tie-averaged) against sklearn's roc_auc_score
    on random data including heavy ties."""
    from sklearn.metrics import roc_auc_score
    rng = np.random.default_rng(0)
    for trial in range(20):
        n = int(rng.integers(10, 500))
        labels = (rng.random(n) > 0.5).astype(np.float64)
        if labels.min() == labels.max():
            labels[0] = 1.0 - labels[0]
        # quantized scores force tie handling
        scores = np.round(rng.random(n), int(rng.integers(0, 3)))
        ours = auc_score(scores, labels)
        ref = roc_auc_score(labels, scores)
        assert abs(ours - ref) < 1e-9, f"trial {trial}: {ours} vs {ref}"
