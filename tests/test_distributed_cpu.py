"""Multi-process data-parallel tests on CPU (gloo, world_size=2).

These validate the RCCL-path semantics (bucketed all-reduce overlapped with
backward, sparse embedding aggregation, window mode) with the gloo backend so
the distributed engine is correct by construction before it ever touches an
8-GPU node (the driver runs the real multi-GPU scaling bench)."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from shifu_amd.data.csv_loader import TabularDataset
from shifu_amd.data.synthetic import synthetic_arrays

WORLD = 2


def _init(rank, world, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _dense_dp_worker(rank, port, q):
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.ops.linear import FusedLinear
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.parallel.dist import GradAggregator
    try:
        _init(rank, WORLD, port)
        torch.manual_seed(0)
        model = torch.nn.Sequential(FusedLinear(6, 8, "relu", seed=1),
                                    FusedLinear(8, 1, "none", seed=2))
        dense, _ = split_params(model)[0], None
        flat = FlatParams(dense)
        agg = GradAggregator(flat, [], bucket_mb=1, overlap=True)

        # per-rank shard: rank 0 gets rows 0..7, rank 1 rows 8..15
        dn, _, tg, _ = synthetic_arrays(16, 6, seed=7, weighted=False)
        x = torch.from_numpy(dn[rank * 8:(rank + 1) * 8])
        y = torch.from_numpy(tg[rank * 8:(rank + 1) * 8])
        w = torch.ones(8)

        logits = model(x).reshape(-1)
        loss = weighted_loss(logits, y, w, "sigmoid_ce")
        loss.backward()
        agg.finish()

        # reference: single-process full-batch grad
        model2 = torch.nn.Sequential(FusedLinear(6, 8, "relu", seed=1),
                                     FusedLinear(8, 1, "none", seed=2))
        xf = torch.from_numpy(dn)
        yf = torch.from_numpy(tg)
        lf = weighted_loss(model2(xf).reshape(-1), yf, torch.ones(16), "sigmoid_ce")
        lf.backward()
        ref = torch.cat([p.grad.reshape(-1) for _, p in
                         sorted(model2.named_parameters(), key=lambda kv: kv[0])])
        ok = torch.allclose(flat.flat_grad, ref, atol=1e-6)
        q.put((rank, bool(ok), float((flat.flat_grad - ref).abs().max())))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _sparse_dp_worker(rank, port, q):
    from shifu_amd.ops.embedding import MultiEmbedding
    from shifu_amd.ops.flat import FlatParams
    from shifu_amd.parallel.dist import GradAggregator
    try:
        _init(rank, WORLD, port)
        emb = MultiEmbedding([10], dim=2, seed=0)
        agg = GradAggregator(FlatParams([]), [emb.arena])
        ids = torch.tensor([[1], [2]]) if rank == 0 else torch.tensor([[2], [3]])
        out = emb(ids)
        out.sum().backward()
        agg.finish()
        g = emb.arena.grad.coalesce().to_dense()
        # averaged across ranks: row1=0.5, row2=1.0, row3=0.5 per dim
        expect = torch.zeros(10, 2)
        expect[1] = 0.5
        expect[2] = 1.0
        expect[3] = 0.5
        q.put((rank, bool(torch.allclose(g, expect, atol=1e-6))))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _trainer_dp_worker(rank, port, tmpdir, q):
    from shifu_amd.config.model_config import ModelConfig
    from shifu_amd.config.run_config import RunConfig
    from shifu_amd.data.sharding import shard_rows
    from shifu_amd.models.mlp import ShifuMLP
    from shifu_amd.train.trainer import Trainer
    try:
        _init(rank, WORLD, port)
        dense, cats, target, weight = synthetic_arrays(600, 6, seed=3)
        full = TabularDataset(dense, cats, target, weight)
        train, valid = full.split(0.2, seed=1)
        s, e = shard_rows(len(train), rank, WORLD)
        shard = train.subset(np.arange(s, e))
        mc = ModelConfig.from_dict({
            "train": {"numTrainEpochs": 2, "params": {
                "NumHiddenLayers": 2, "NumHiddenNodes": [16, 8],
                "ActivationFunc": ["relu", "relu"], "LearningRate": 0.02,
                "Optimizer": "adam", "Loss": "sigmoid_ce",
                "MiniBatchSize": 50, "L2Reg": 0.0}}})
        rc = RunConfig(num_gpus=WORLD, tmp_model_path=os.path.join(tmpdir, "ckpt"),
                       final_model_path=os.path.join(tmpdir, "final"))
        model = ShifuMLP(6, [16, 8], ["relu", "relu"], seed=4)
        tr = Trainer(model, mc, rc, shard, valid, rank=rank, world_size=WORLD)
        tr.fit()
        # every rank must end with IDENTICAL parameters
        flat_sum = float(tr.flat.flat.sum())
        q.put((rank, flat_sum))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run_workers(fn, port, extra=()):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, port) + tuple(extra) + (q,))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=180) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0, f"worker exit {p.exitcode}"
    return outs


def test_dense_allreduce_matches_fullbatch():
    outs = _run_workers(_dense_dp_worker, 29611)
    for rank, ok, maxdiff in outs:
        assert ok, f"rank {rank} grad mismatch {maxdiff}"


def test_sparse_embedding_aggregation():
    outs = _run_workers(_sparse_dp_worker, 29613)
    for rank, ok in outs:
        assert ok, f"rank {rank} sparse grad mismatch"


def test_trainer_2rank_params_converge(tmp_path):
    outs = _run_workers(_trainer_dp_worker, 29615, extra=(str(tmp_path),))
    sums = [s for _, s in outs]
    assert abs(sums[0] - sums[1]) < 1e-5, f"ranks diverged: {sums}"


def test_trainer_2rank_run_to_run_deterministic(tmp_path):
    """Two identical 2-rank runs must produce BITWISE identical final
    parameters (gloo all-reduce + fp32 CPU path are deterministic)."""
    o1 = _run_workers(_trainer_dp_worker, 29617, extra=(str(tmp_path / "a"),))
    o2 = _run_workers(_trainer_dp_worker, 29619, extra=(str(tmp_path / "b"),))
    assert sorted(o1) == sorted(o2), f"cross-run divergence: {o1} vs {o2}"
