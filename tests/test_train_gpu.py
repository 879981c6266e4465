"""GPU end-to-end correctness stress: full-model parity vs the CPU reference
path, window-mode training, and checkpoint-resume equivalence."""
import os
import tempfile

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from shifu_amd.config.model_config import ModelConfig
from shifu_amd.config.run_config import RunConfig
from shifu_amd.data.csv_loader import TabularDataset
from shifu_amd.data.synthetic import synthetic_arrays
from shifu_amd.models.deepfm import DeepFM
from shifu_amd.models.mlp import ShifuMLP
from shifu_amd.models.wide_deep import WideDeep
from shifu_amd.train.trainer import Trainer


def _mc(epochs=2, batch=256, window=1, opt="adam", lr=0.01):
    return ModelConfig.from_dict({"train": {"numTrainEpochs": epochs, "params": {
        "NumHiddenLayers": 2, "NumHiddenNodes": [64, 32],
        "ActivationFunc": ["relu", "tanh"], "LearningRate": lr,
        "Optimizer": opt, "Loss": "sigmoid_ce", "MiniBatchSize": batch,
        "L2Reg": 0.01, "UpdateWindow": window}}})


@pytest.mark.parametrize("cls,vocab", [(WideDeep, (500, 300)), (DeepFM, (400, 400)),
                                       (ShifuMLP, ())])
def test_full_model_fwd_parity_gpu_vs_cpu(cls, vocab):
    """Whole-model forward through the HIP bf16 path vs the fp32 CPU
    reference on identical weights — catches composition-level numerics
    drift no per-kernel test sees."""
    torch.manual_seed(0)
    if cls is ShifuMLP:
        m_cpu = ShifuMLP(24, [64, 32], ["relu", "tanh"], seed=3)
    else:
        m_cpu = cls(24, list(vocab), 8, [64, 32], ["relu", "tanh"], seed=3)
    import copy
    from shifu_amd.ops.embedding import MultiEmbedding
    m_gpu = copy.deepcopy(m_cpu).cuda()
    for mod in m_gpu.modules():
        if isinstance(mod, MultiEmbedding):   # deepcopy drops tensor attrs
            mod.arena.data = mod.arena.data.to(torch.bfloat16)
            mod.arena._is_embedding_arena = True

    g = torch.Generator().manual_seed(5)
    dense = torch.randn(512, 24, generator=g)
    cats = (torch.stack([torch.randint(0, v, (512,), generator=g) for v in vocab], 1)
            if vocab else None)
    if cls is ShifuMLP:
        out_cpu = m_cpu(dense)
        out_gpu = m_gpu(dense.cuda().to(torch.bfloat16))
    else:
        out_cpu = m_cpu(dense, cats)
        out_gpu = m_gpu(dense.cuda().to(torch.bfloat16), cats.cuda())
    a, b = out_gpu.float().cpu(), out_cpu.float()
    rel = float((a - b).abs().max() / b.abs().max().clamp_min(1e-2))
    assert rel < 0.08, f"{cls.__name__} fwd drift rel={rel}"


def test_window_mode_gpu_learns():
    dense, cats, target, weight = synthetic_arrays(4096, 12, (200, 200), seed=5)
    full = TabularDataset(dense, cats, target, weight)
    train, valid = full.split(0.2, seed=1)
    with tempfile.TemporaryDirectory() as td:
        rc = RunConfig(tmp_model_path=td + "/c", final_model_path=td + "/f")
        tr = Trainer(WideDeep(12, [200, 200], 8, [64, 32], ["relu", "tanh"], seed=2),
                     _mc(epochs=3, window=4), rc, train, valid,
                     device=torch.device("cuda"))
        first = tr.evaluate(tr.valid_data)
        tr.fit()
        last = tr.evaluate(tr.valid_data)
        assert last["loss"] < first["loss"]


def test_resume_equivalence_gpu():
    """Train 4 epochs straight vs 2 epochs + resume + 2 epochs: the resumed
    run must land at the same valid loss (same data order via saved RNG)."""
    dense, cats, target, weight = synthetic_arrays(2048, 10, (), seed=9)
    full = TabularDataset(dense, cats, target, weight)
    train, valid = full.split(0.2, seed=1)

    def run(epochs, tmp, resume_from=None):
        rc = RunConfig(tmp_model_path=tmp, final_model_path=tmp + "_f")
        tr = Trainer(ShifuMLP(10, [32, 16], ["relu", "relu"], seed=4),
                     _mc(epochs=epochs, batch=128), rc, train, valid,
                     device=torch.device("cuda"))
        tr.fit()
        return tr.evaluate(tr.valid_data)["loss"]

    with tempfile.TemporaryDirectory() as td:
        straight = run(4, td + "/a")
        run(2, td + "/b")            # writes ckpt-0, ckpt-1
        resumed = run(4, td + "/b")  # resumes at epoch 2
        # split-K wgrad uses f32 atomics, so two runs of the SAME schedule
        # differ by reduction-order rounding; the gate is trajectory
        # equivalence, not bitwise equality
        assert abs(straight - resumed) < 3e-2, (straight, resumed)


def test_adadelta_reference_default_gpu():
    """The reference's default optimizer (Adadelta lr from ModelConfig) runs
    the fused HIP path end-to-end and optimizes."""
    dense, cats, target, weight = synthetic_arrays(4096, 16, (), seed=3)
    full = TabularDataset(dense, cats, target, weight)
    train, valid = full.split(0.2, seed=1)
    with tempfile.TemporaryDirectory() as td:
        rc = RunConfig(tmp_model_path=td + "/c", final_model_path=td + "/f")
        tr = Trainer(ShifuMLP(16, [64, 32], ["relu", "relu"], seed=1),
                     _mc(epochs=4, opt="adadelta", lr=1.0), rc, train, valid,
                     device=torch.device("cuda"))
        first = tr.evaluate(tr.valid_data)
        tr.fit()
        assert tr.evaluate(tr.valid_data)["loss"] < first["loss"]


def test_graphed_trainer_matches_eager():
    """Graph-captured epochs must train identically to eager epochs (the
    capture snapshot/restore must leave no side effects)."""
    dense, cats, target, weight = synthetic_arrays(4096, 12, (300, 300), seed=7)
    full = TabularDataset(dense, cats, target, weight)
    train, valid = full.split(0.2, seed=1)

    def run(graphs):
        with tempfile.TemporaryDirectory() as td:
            rc = RunConfig(tmp_model_path=td + "/c", final_model_path=td + "/f",
                           graphs=graphs)
            tr = Trainer(WideDeep(12, [300, 300], 8, [64, 32], ["relu", "tanh"],
                                  seed=6), _mc(epochs=2, batch=512), rc,
                         train, valid, device=torch.device("cuda"))
            tr.fit()
            assert (tr._graph is not None) == (graphs == "on"), \
                f"graph mode not exercised as expected ({graphs})"
            return tr.evaluate(tr.valid_data)["loss"]

    eager = run("off")
    graphed = run("on")
    assert abs(eager - graphed) < 3e-2, (eager, graphed)


@pytest.mark.parametrize("cfg", [
    dict(loss="weighted_mse", opt="adadelta", lr=1.0, acts=["tanh", "tanh"], batch=333),
    dict(loss="sigmoid_ce", opt="adagrad", lr=0.05, acts=["leakyrelu", "sigmoid"], batch=100),
    dict(loss="weighted_mse", opt="sgd", lr=0.5, acts=["relu", "relu"], batch=1024),
])
def test_config_fuzz_gpu(cfg):
    """Odd batch sizes x every activation/loss/optimizer family through the
    full HIP Trainer: must run, stay finite, and not regress the loss."""
    dense, cats, target, weight = synthetic_arrays(3000, 9, (123, 77), seed=13)
    full = TabularDataset(dense, cats, target, weight)
    train, valid = full.split(0.2, seed=1)
    mc = ModelConfig.from_dict({"train": {"numTrainEpochs": 2, "params": {
        "NumHiddenLayers": 2, "NumHiddenNodes": [48, 24],
        "ActivationFunc": cfg["acts"], "LearningRate": cfg["lr"],
        "Optimizer": cfg["opt"], "Loss": cfg["loss"],
        "MiniBatchSize": cfg["batch"], "L2Reg": 0.05}}})
    with tempfile.TemporaryDirectory() as td:
        rc = RunConfig(tmp_model_path=td + "/c", final_model_path=td + "/f")
        tr = Trainer(WideDeep(9, [123, 77], 8, [48, 24], cfg["acts"], seed=3),
                     mc, rc, train, valid, device=torch.device("cuda"))
        first = tr.evaluate(tr.valid_data)
        tr.fit()
        last = tr.evaluate(tr.valid_data)
        assert np.isfinite(last["loss"])
        # correctness gate is "no blow-up" — short Adadelta/wMSE runs may
        # wobble a few % before descending
        assert last["loss"] <= first["loss"] * 1.25


def test_streaming_residency_gpu():
    """Host-pinned streaming residency trains like the HBM-resident path
    (dense MLP: fully deterministic kernels -> tight tolerance)."""
    dense, cats, target, weight = synthetic_arrays(2000, 9, (), seed=17)
    full = TabularDataset(dense, cats, target, weight)
    train, valid = full.split(0.2, seed=1)
    mc = ModelConfig.from_dict({"train": {"numTrainEpochs": 2, "params": {
        "NumHiddenLayers": 2, "NumHiddenNodes": [48, 24],
        "ActivationFunc": ["relu", "relu"], "LearningRate": 0.01,
        "Optimizer": "adam", "Loss": "sigmoid_ce",
        "MiniBatchSize": 256, "L2Reg": 0.0}}})

    def run(residency, td):
        rc = RunConfig(tmp_model_path=td + "/c", final_model_path=td + "/f",
                       data_residency=residency, graphs="off")
        tr = Trainer(ShifuMLP(9, [48, 24], ["relu", "relu"], seed=3), mc, rc,
                     train, valid, device=torch.device("cuda"))
        tr.fit()
        return tr.evaluate(tr.valid_data)["loss"]

    with tempfile.TemporaryDirectory() as td:
        resident = run("auto", td + "/a")
        streamed = run("stream", td + "/b")
    assert np.isfinite(streamed)
    assert abs(resident - streamed) < 1e-3, (resident, streamed)


def test_rccl_op_smoke_world1():
    """Exercise every (collective, dtype) combo the multi-GPU step uses on
    the REAL RCCL backend (world=1): bucketed fp32 all_reduce, int32 and
    bf16 all_to_all_single with explicit splits, bf16 all_gather, int64
    all_reduce MAX (uniform stepping).  First hardware contact for these
    paths otherwise happens in the driver's 8-GPU SCALE run."""
    import os
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29771")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        dev = torch.device("cuda", 0)
        g = torch.randn(1 << 20, device=dev)
        ref = g.clone()
        dist.all_reduce(g, op=dist.ReduceOp.SUM)
        assert torch.equal(g, ref)

        ids = torch.randint(0, 1000, (64, 32), dtype=torch.int32, device=dev)
        out = torch.empty_like(ids)
        dist.all_to_all_single(out, ids, [64], [64])
        assert torch.equal(out, ids)

        vals = torch.randn(512, 66, device=dev).to(torch.bfloat16)
        vout = torch.empty_like(vals)
        dist.all_to_all_single(vout, vals, [512], [512])
        assert torch.equal(vout, vals)

        pad = torch.randn(128, 8, device=dev).to(torch.bfloat16)
        outs = [torch.empty_like(pad)]
        dist.all_gather(outs, pad)
        assert torch.equal(outs[0], pad)

        t = torch.tensor([7], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        assert int(t) == 7
    finally:
        dist.destroy_process_group()


def test_hipgraph_captures_rccl_collectives():
    """hipGraph capture of a full table-EP training step INCLUDING the RCCL
    collectives (all_to_all_single + bucketed all_reduce) at world=1 — the
    mechanism proof for enabling graphs at world>1 (tools/graph_rccl_test)."""
    import subprocess
    import sys
    out = subprocess.run([sys.executable, "tools/graph_rccl_test.py"],
                         capture_output=True, text=True, timeout=300,
                         cwd=os.path.dirname(os.path.dirname(
                             os.path.abspath(__file__))))
    assert "GRAPH_RCCL_OK" in out.stdout, out.stderr[-1500:]
