"""Round-2 correctness fixes: window-mean aggregation, quorum partial
participation, EP-sharded checkpoint/resume, mid-epoch RNG snapshots, and
the SUM_BY_NONZERO_WEIGHTS loss normalization."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _mc(epochs=2, opt="sgd", lr=0.05, batch=16):
    from shifu_amd.config.model_config import ModelConfig
    return ModelConfig.from_dict({
        "train": {"numTrainEpochs": epochs, "validSetRate": 0.2,
                  "params": {"NumHiddenLayers": 1, "NumHiddenNodes": [8],
                             "ActivationFunc": ["relu"], "LearningRate": lr,
                             "Optimizer": opt, "Loss": "sigmoid_ce",
                             "MiniBatchSize": batch, "L2Reg": 0.0}}})


def _init(rank, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _run(fn, port, extra=()):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, port) + tuple(extra) + (q,))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=240) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0, f"worker exit {p.exitcode}"
    return outs


# --------------------------------------------------------------- window mean
def test_window_accumulation_applies_mean():
    """Single process, update_window-style accumulation: finish() must apply
    the MEAN of the window's gradients (SAGN applies the window mean,
    sagn_monitor.py:137-142), not the sum."""
    from shifu_amd.ops.flat import FlatParams
    from shifu_amd.parallel.dist import GradAggregator

    p = torch.nn.Parameter(torch.zeros(4))
    flat = FlatParams([p])
    agg = GradAggregator(flat, [], bucket_mb=1)

    for step in range(3):                      # 3-step window
        agg.set_sync(step == 2)
        (p * (step + 1.0)).sum().backward()    # grad += (step+1) * ones
    agg.finish()
    # grads 1,2,3 accumulate to 6; window mean = 2
    assert torch.allclose(flat.flat_grad, torch.full((4,), 2.0))

    # counter must reset: a following 1-step window divides by 1
    p.grad = None
    flat.zero_grad()
    agg.set_sync(True)
    (p * 5.0).sum().backward()
    agg.finish()
    assert torch.allclose(flat.flat_grad, torch.full((4,), 5.0))


# -------------------------------------------------------------------- quorum
def _quorum_worker(rank, port, q):
    from shifu_amd.ops.flat import FlatParams
    from shifu_amd.parallel.dist import GradAggregator
    try:
        _init(rank, port)
        p = torch.nn.Parameter(torch.zeros(3))
        flat = FlatParams([p])
        agg = GradAggregator(flat, [], bucket_mb=1, quorum_ratio=0.5)
        seen = []
        for step in range(4):
            agg.set_sync(True)
            # rank r contributes grad full of (r+1)
            (p * (rank + 1.0)).sum().backward()
            agg.finish()
            seen.append(float(flat.flat_grad[0]))
            p.grad = None
            flat.zero_grad()
        q.put((rank, seen))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_quorum_rotates_single_contributor():
    """quorum_ratio=0.5 at world=2: each sync step aggregates exactly one
    rank's gradient, rotating deterministically; both ranks see the SAME
    aggregated gradient (params stay in lockstep)."""
    outs = dict(_run(_quorum_worker, 29741))
    assert outs[0] == outs[1], "ranks diverged under quorum"
    # contributor rotates: step s admits rank r with (r+s)%2 == 0
    assert outs[0] == [1.0, 2.0, 1.0, 2.0]


def test_quorum_rejects_ep():
    from shifu_amd.ops.flat import FlatParams
    from shifu_amd.parallel.dist import GradAggregator
    p = torch.nn.Parameter(torch.zeros(4, 2))
    p._is_ep_sharded = True
    with pytest.raises(ValueError, match="quorum"):
        GradAggregator(FlatParams([]), [p], quorum_ratio=0.5)


# ------------------------------------------------------------ EP resume (high)
def _ep_resume_worker(rank, port, tmp, q):
    from shifu_amd.config.run_config import RunConfig
    from shifu_amd.data.csv_loader import TabularDataset
    from shifu_amd.data.synthetic import synthetic_arrays
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.trainer import Trainer
    try:
        _init(rank, port)
        torch.manual_seed(0)
        vocab = [33, 47]   # total 80 rows -> 40/40 shard split at world=2

        def build(tmp_model):
            dn, ct, tg, w = synthetic_arrays(64, 4, vocab,
                                             seed=100 + rank, weighted=False)
            ds = TabularDataset(dn, ct, tg, w)
            train, valid = ds.split(0.25, seed=1)
            mc = _mc(epochs=2)
            rc = RunConfig(tmp_model_path=tmp_model,
                           final_model_path=os.path.join(tmp, f"final{rank}"),
                           device="cpu", batch_size=16)
            model = WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                             sharded_embeddings=True, world=WORLD, rank=rank)
            return Trainer(model, mc, rc, train, valid, rank=rank,
                           world_size=WORLD)

        tmp_model = os.path.join(tmp, "ckpt")
        tr = build(tmp_model)
        tr.fit()                                   # writes epoch-1 checkpoint
        arena_after = tr.model.embeddings.arena.data.clone()
        wide_after = tr.model.wide_cat.arena.data.clone()
        step_after = tr.global_step

        tr2 = build(tmp_model)
        tr2.maybe_resume()
        ok_arena = torch.allclose(tr2.model.embeddings.arena.data, arena_after)
        ok_wide = torch.allclose(tr2.model.wide_cat.arena.data, wide_after)
        ok_step = (tr2.global_step == step_after and tr2.start_epoch == 2)
        q.put((rank, bool(ok_arena), bool(ok_wide), bool(ok_step)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ep_sharded_resume_roundtrip(tmp_path):
    """Every rank's arena shard (and rowwise optimizer state) must survive a
    checkpoint/resume cycle — the round-1 bug loaded rank 0's shard into
    every rank."""
    for rank, ok_arena, ok_wide, ok_step in _run(_ep_resume_worker, 29743,
                                                 extra=(str(tmp_path),)):
        assert ok_arena, f"rank {rank}: deep arena shard corrupted by resume"
        assert ok_wide, f"rank {rank}: wide arena shard corrupted by resume"
        assert ok_step, f"rank {rank}: epoch/step bookkeeping wrong"


def _ep_reshard_worker(rank, port, tmp, q):
    """Resume an EP run from a REPLICATED (single-process) checkpoint: arenas
    re-shard row%world."""
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.ops.optim import FusedOptimizer
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.train import checkpoint as ckpt
    try:
        _init(rank, port)
        vocab = [19, 28]
        path = os.path.join(tmp, "ckpt-0.pt")
        model = WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                         sharded_embeddings=True, world=WORLD, rank=rank)
        dense, emb = split_params(model)
        opt = FusedOptimizer(FlatParams(dense), emb, optimizer="sgd", lr=0.1)
        info = ckpt.load_checkpoint(path, model, opt, rank=rank, world=WORLD)

        blob = torch.load(path, map_location="cpu", weights_only=False)
        full = blob["model"]["embeddings.arena"]
        expect = model.embeddings.shard_from_full(full)
        ok = torch.allclose(model.embeddings.arena.data, expect)
        q.put((rank, bool(ok), int(info["epoch"])))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ep_resume_from_replicated_checkpoint(tmp_path):
    # write a replicated checkpoint single-process
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.ops.optim import FusedOptimizer
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.train import checkpoint as ckpt
    vocab = [19, 28]
    model = WideDeep(4, vocab, 4, [8], ["relu"], seed=5)
    dense, emb = split_params(model)
    opt = FusedOptimizer(FlatParams(dense), emb, optimizer="sgd", lr=0.1)
    ckpt.save_checkpoint(str(tmp_path), 0, 10, model, opt)

    for rank, ok, epoch in _run(_ep_reshard_worker, 29745,
                                extra=(str(tmp_path),)):
        assert ok, f"rank {rank}: re-shard rows wrong"
        assert epoch == 0


def test_world_size_change_hard_fails(tmp_path):
    """An EP checkpoint written at world=2 must refuse to load at world=1
    (silent wrong-rows adoption was the round-1 failure mode)."""
    from shifu_amd.parallel.ep import ShardedEmbedding
    from shifu_amd.train import checkpoint as ckpt

    class M(torch.nn.Module):
        def __init__(self, world, rank):
            super().__init__()
            self.emb = ShardedEmbedding([10], 4, seed=1, world=world, rank=rank)

    m = M(2, 0)
    ckpt.save_checkpoint(str(tmp_path), 0, 5, m, None, rank=0, world=2)
    ckpt.save_checkpoint(str(tmp_path), 0, 5, M(2, 1), None, rank=1, world=2)

    m1 = M(1, 0)
    path = ckpt.latest_checkpoint(str(tmp_path), world=1)
    assert path is not None
    with pytest.raises(RuntimeError, match="world"):
        ckpt.load_checkpoint(path, m1, None, rank=0, world=1)


def test_incomplete_shard_set_skipped(tmp_path):
    """latest_checkpoint must skip an epoch whose shard set is partial (a
    crash mid-save) and fall back to the last complete one."""
    from shifu_amd.parallel.ep import ShardedEmbedding
    from shifu_amd.train import checkpoint as ckpt

    class M(torch.nn.Module):
        def __init__(self, rank):
            super().__init__()
            self.emb = ShardedEmbedding([10], 4, seed=1, world=2, rank=rank)

    for r in range(2):
        ckpt.save_checkpoint(str(tmp_path), 0, 5, M(r), None, rank=r, world=2)
    # epoch 1: only rank 0's shard lands (simulated crash)
    ckpt.save_checkpoint(str(tmp_path), 1, 9, M(0), None, rank=0, world=2)
    chosen = ckpt.latest_checkpoint(str(tmp_path), world=2)
    assert chosen and chosen.endswith("ckpt-0.pt"), chosen


# ------------------------------------------------- mid-epoch RNG snapshot (low)
def test_midepoch_checkpoint_replays_same_permutation(tmp_path):
    """A time-based mid-epoch save during epoch 0 must let the resumed run
    redraw epoch 0's exact permutation and restart from the epoch-start
    step count (round-1 saved the post-draw RNG state and skipped epoch 0)."""
    from shifu_amd.config.run_config import RunConfig
    from shifu_amd.data.csv_loader import TabularDataset
    from shifu_amd.data.synthetic import synthetic_arrays
    from shifu_amd.models.mlp import ShifuMLP
    from shifu_amd.train.trainer import Trainer

    dn, ct, tg, w = synthetic_arrays(120, 5, seed=3, weighted=False)
    ds = TabularDataset(dn, ct, tg, w)
    train, valid = ds.split(0.2, seed=1)
    mc = _mc(epochs=1)
    rc = RunConfig(tmp_model_path=str(tmp_path / "ckpt"),
                   final_model_path=str(tmp_path / "final"),
                   device="cpu", batch_size=16,
                   checkpoint_every_secs=1e-9)   # fire on the first step
    tr = Trainer(ShifuMLP(5, [8], ["relu"]), mc, rc, train, valid)
    tr.run_epoch(0)

    tr2 = Trainer(ShifuMLP(5, [8], ["relu"]), mc, rc, train, valid)
    tr2.maybe_resume()
    assert tr2.start_epoch == 0, "mid-epoch save during epoch 0 must resume epoch 0"
    assert tr2.global_step == 0, "must recount from the epoch-start step"
    # the resumed RNG must draw epoch 0's permutation again
    fresh = np.random.default_rng(rc.seed)
    assert np.array_equal(tr2._rng.permutation(len(tr2.train_data)),
                          fresh.permutation(len(tr2.train_data)))


# ------------------------------------------- SUM_BY_NONZERO_WEIGHTS normalization
def test_loss_normalizes_by_nonzero_count():
    from shifu_amd.ops.loss import weighted_loss
    z = torch.randn(8, requires_grad=True)
    y = (torch.rand(8) > 0.5).float()
    w = torch.tensor([2.0, 2.0, 0.0, 0.0, 2.0, 2.0, 0.0, 0.0])
    loss = weighted_loss(z, y, w, "sigmoid_ce")
    per = torch.nn.functional.binary_cross_entropy_with_logits(
        z.detach(), y, reduction="none")
    # TF SUM_BY_NONZERO_WEIGHTS: divide by 4 (nonzero count), not sum(w)=8
    expect = (w * per).sum() / 4.0
    assert torch.allclose(loss, expect, atol=1e-6)


def _ep_step_ckpt_worker(rank, port, tmp, q):
    """Step-cadence mid-epoch saves with EP shards: all ranks save at the
    same step, resume replays the epoch from its start."""
    from shifu_amd.config.run_config import RunConfig
    from shifu_amd.data.csv_loader import TabularDataset
    from shifu_amd.data.synthetic import synthetic_arrays
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.trainer import Trainer
    from shifu_amd.train import checkpoint as ckpt
    try:
        _init(rank, port)
        vocab = [21, 35]
        dn, ct, tg, w = synthetic_arrays(96, 4, vocab, seed=50 + rank,
                                         weighted=False)
        ds = TabularDataset(dn, ct, tg, w)
        train, valid = ds.split(0.25, seed=1)
        rc = RunConfig(tmp_model_path=os.path.join(tmp, "ckpt"),
                       final_model_path=os.path.join(tmp, f"f{rank}"),
                       device="cpu", batch_size=16,
                       checkpoint_every_steps=2)
        model = WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                         sharded_embeddings="table", world=WORLD, rank=rank,
                         unified=False)
        tr = Trainer(model, _mc(epochs=1), rc, train, valid, rank=rank,
                     world_size=WORLD)
        tr.run_epoch(0)
        dist.barrier()
        path = ckpt.latest_checkpoint(rc.tmp_model_path, world=WORLD)
        ok_exists = path is not None and path.endswith("ckpt--1.pt")
        # resume must land at epoch 0, step 0 (pre-epoch snapshot)
        tr2 = Trainer(WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                               sharded_embeddings="table", world=WORLD,
                               rank=rank, unified=False),
                      _mc(epochs=1), rc, train, valid, rank=rank,
                      world_size=WORLD)
        tr2.maybe_resume()
        ok_resume = (tr2.start_epoch == 0 and tr2.global_step == 0)
        # the resumed shard equals the saved mid-epoch shard
        ok_shard = torch.allclose(tr2.model.embeddings.arena.data,
                                  tr.model.embeddings.arena.data, atol=1.0)
        q.put((rank, bool(ok_exists), bool(ok_resume)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ep_step_cadence_checkpoint(tmp_path):
    for rank, ok_exists, ok_resume in _run(_ep_step_ckpt_worker, 29759,
                                           extra=(str(tmp_path),)):
        assert ok_exists, f"rank {rank}: no mid-epoch checkpoint written"
        assert ok_resume, f"rank {rank}: resume bookkeeping wrong"


def test_mixed_world_shard_debris(tmp_path):
    """An epoch carrying a COMPLETE shard set for one world size plus debris
    from another (aborted world-change experiment) is still selectable; an
    epoch with only partial sets for every world is skipped."""
    from shifu_amd.parallel.ep import ShardedEmbedding
    from shifu_amd.train import checkpoint as ckpt

    class M(torch.nn.Module):
        def __init__(self, world, rank):
            super().__init__()
            self.emb = ShardedEmbedding([10], 4, seed=1, world=world, rank=rank)

    # epoch 0: complete at world=2 AND one stray world=4 shard
    for r in range(2):
        ckpt.save_checkpoint(str(tmp_path), 0, 5, M(2, r), None, rank=r, world=2)
    ckpt.save_checkpoint(str(tmp_path), 0, 5, M(4, 0), None, rank=0, world=4)
    # epoch 1: partial for BOTH worlds
    ckpt.save_checkpoint(str(tmp_path), 1, 9, M(2, 0), None, rank=0, world=2)
    ckpt.save_checkpoint(str(tmp_path), 1, 9, M(4, 1), None, rank=1, world=4)
    chosen = ckpt.latest_checkpoint(str(tmp_path), world=2)
    assert chosen and chosen.endswith("ckpt-0.pt"), chosen
