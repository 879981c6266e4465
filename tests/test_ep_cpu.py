"""Sharded-embedding (EP, all-to-all) correctness on CPU/gloo world_size=2:
forward parity with the replicated MultiEmbedding and gradient parity with
the DP sparse-allgather path."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _init(rank, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _ep_worker(rank, port, q):
    from shifu_amd.ops.embedding import MultiEmbedding
    from shifu_amd.parallel.ep import ShardedEmbedding
    try:
        _init(rank, port)
        torch.manual_seed(0)
        vocab = [11, 23]
        ref = MultiEmbedding(vocab, dim=4, seed=7)          # replicated twin
        ep = ShardedEmbedding(vocab, dim=4, seed=7, world=WORLD, rank=rank)

        # shard init parity: ep rows == ref arena rows rank::world
        ok_init = torch.allclose(ep.arena.data, ref.arena.data[rank::WORLD])

        g = torch.Generator().manual_seed(3)
        ids = torch.randint(0, 11, (6, 2), generator=g)
        ids[:, 1] = torch.randint(0, 23, (6,), generator=g)

        out_ep = ep(ids)
        out_ref = ref(ids)
        ok_fwd = torch.allclose(out_ep, out_ref, atol=1e-6)

        # backward: weighted sum loss; each rank uses a DIFFERENT ids batch
        ids_r = ids + rank  # different rows per rank (clamped in forward)
        out = ep(ids_r)
        (out * (rank + 1.0)).sum().backward()
        grad = ep.arena.grad
        ok_sparse = grad is not None and grad.is_sparse

        # reference: replicated arenas on both ranks with the same combined
        # gradient: sum over ranks of per-rank grads / world
        ref2 = MultiEmbedding(vocab, dim=4, seed=7)
        for r in range(WORLD):
            o = ref2(ids + r)
            (o * (r + 1.0)).sum().backward()
        expected_full = ref2.arena.grad.coalesce().to_dense() / WORLD
        got = torch.zeros_like(expected_full[rank::WORLD])
        from shifu_amd.ops.embedding import sparse_rows_values
        rows, vals = sparse_rows_values(grad)
        got.index_add_(0, rows, vals.to(got.dtype))
        ok_bwd = torch.allclose(got, expected_full[rank::WORLD], atol=1e-5)

        q.put((rank, bool(ok_init), bool(ok_fwd), bool(ok_sparse), bool(ok_bwd)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _ep_train_worker(rank, port, mode, q):
    """Full WideDeep training-step parity: EP model step == DP model step."""
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.ops.optim import FusedOptimizer
    from shifu_amd.parallel.dist import GradAggregator
    try:
        _init(rank, port)
        torch.manual_seed(0)
        vocab = [50, 70]

        def build(sharded):
            return WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                            sharded_embeddings=sharded, world=WORLD, rank=rank)

        def one_step(model):
            dense_params, emb_params = split_params(model)
            flat = FlatParams(dense_params)
            agg = GradAggregator(flat, emb_params, bucket_mb=1)
            opt = FusedOptimizer(flat, emb_params, optimizer="sgd", lr=0.1,
                                 l2_reg=0.0, emb_optimizer="sgd", emb_lr=0.1)
            g = torch.Generator().manual_seed(100 + rank)
            dense = torch.randn(8, 4, generator=g)
            cats = torch.randint(0, 50, (8, 2), generator=g)
            y = (torch.rand(8, generator=g) > 0.5).float()
            w = torch.ones(8)
            loss = weighted_loss(model(dense, cats), y, w, "sigmoid_ce")
            loss.backward()
            agg.finish()
            opt.step()
            return model

        # EP model and DP model take one identical step
        m_ep = one_step(build(mode))
        m_dp = one_step(build(False))

        # dense towers must match exactly
        ok_dense = all(torch.allclose(a.detach(), b.detach(), atol=1e-5)
                       for (na, a), (nb, b) in zip(
                           sorted(m_ep.named_parameters(), key=lambda kv: kv[0]),
                           sorted(m_dp.named_parameters(), key=lambda kv: kv[0]))
                       if "arena" not in na)
        # EP shard must equal its slice of the DP arena after the update
        # (shard_from_full knows the topology: row%world or by-feature)
        ok_emb = torch.allclose(
            m_ep.embeddings.arena.data,
            m_ep.embeddings.shard_from_full(m_dp.embeddings.arena.data),
            atol=1e-5)
        q.put((rank, bool(ok_dense), bool(ok_emb)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run(fn, port, extra=()):
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


def test_ep_forward_backward_parity():
    for rank, ok_init, ok_fwd, ok_sparse, ok_bwd in _run(_ep_worker, 29721):
        assert ok_init, f"rank {rank}: shard init mismatch"
        assert ok_fwd, f"rank {rank}: forward mismatch"
        assert ok_sparse, f"rank {rank}: grad not sparse"
        assert ok_bwd, f"rank {rank}: backward grad mismatch"


@pytest.mark.parametrize("mode,port", [("row", 29723), ("table", 29733)])
def test_ep_training_step_matches_dp(mode, port):
    for rank, ok_dense, ok_emb in _run(_ep_train_worker, port, extra=(mode,)):
        assert ok_dense, f"rank {rank}: dense params diverged ({mode})"
        assert ok_emb, f"rank {rank}: embedding shard diverged from DP ({mode})"


def _ep_pair_worker(rank, port, q):
    """Shared-routing pair gather == two independent ShardedEmbedding lookups."""
    from shifu_amd.parallel.ep import ShardedEmbedding, ep_pair_gather
    try:
        _init(rank, port)
        vocab = [13, 29]
        e1 = ShardedEmbedding(vocab, dim=1, seed=4, world=WORLD, rank=rank)
        e2 = ShardedEmbedding(vocab, dim=6, seed=9, world=WORLD, rank=rank)
        g = torch.Generator().manual_seed(8 + rank)
        ids = torch.stack([torch.randint(0, 13, (10,), generator=g),
                           torch.randint(0, 29, (10,), generator=g)], dim=1)
        o1p, o2p = ep_pair_gather(e1, e2, ids)
        o1, o2 = e1(ids), e2(ids)
        ok_fwd = torch.allclose(o1p, o1, atol=1e-6) and torch.allclose(o2p, o2, atol=1e-6)

        # backward parity: same loss through both paths gives same arena grads
        (o1p.sum() * 2 + o2p.pow(2).sum()).backward()
        g1p = e1.arena.grad.coalesce().to_dense().clone()
        g2p = e2.arena.grad.coalesce().to_dense().clone()
        e1.arena.grad = None
        e2.arena.grad = None
        o1b, o2b = e1(ids), e2(ids)
        (o1b.sum() * 2 + o2b.pow(2).sum()).backward()
        ok_bwd = (torch.allclose(g1p, e1.arena.grad.coalesce().to_dense(), atol=1e-5) and
                  torch.allclose(g2p, e2.arena.grad.coalesce().to_dense(), atol=1e-5))
        q.put((rank, bool(ok_fwd), bool(ok_bwd)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ep_pair_gather_shared_routing():
    for rank, ok_fwd, ok_bwd in _run(_ep_pair_worker, 29725):
        assert ok_fwd, f"rank {rank}: pair forward mismatch"
        assert ok_bwd, f"rank {rank}: pair backward mismatch"


def _ep_window_worker(rank, port, mode, q):
    """Window (local-SGD) mode with EP arenas: 4 steps syncing every 2nd must
    produce the same model as the replicated-DP path — sparse grads accumulate
    across the window on both paths."""
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.ops.optim import FusedOptimizer
    from shifu_amd.parallel.dist import GradAggregator
    try:
        _init(rank, port)
        torch.manual_seed(0)
        vocab = [50, 70]

        def run(sharded):
            model = WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                             sharded_embeddings=sharded, world=WORLD, rank=rank)
            dense_params, emb_params = split_params(model)
            flat = FlatParams(dense_params)
            agg = GradAggregator(flat, emb_params, bucket_mb=1)
            opt = FusedOptimizer(flat, emb_params, optimizer="sgd", lr=0.05,
                                 l2_reg=0.0, emb_optimizer="sgd", emb_lr=0.05)
            for si in range(4):
                sync = (si % 2 == 1)
                agg.set_sync(sync)
                g = torch.Generator().manual_seed(200 + 10 * rank + si)
                dense = torch.randn(8, 4, generator=g)
                cats = torch.randint(0, 50, (8, 2), generator=g)
                y = (torch.rand(8, generator=g) > 0.5).float()
                w = torch.ones(8)
                loss = weighted_loss(model(dense, cats), y, w, "sigmoid_ce")
                loss.backward()
                if sync:
                    agg.finish()
                    opt.step()
                    opt.zero_grad()
            return model

        m_ep = run(mode)
        m_dp = run(False)
        ok_dense = all(torch.allclose(a.detach(), b.detach(), atol=1e-5)
                       for (na, a), (nb, b) in zip(
                           sorted(m_ep.named_parameters(), key=lambda kv: kv[0]),
                           sorted(m_dp.named_parameters(), key=lambda kv: kv[0]))
                       if "arena" not in na)
        ok_emb = (torch.allclose(
                      m_ep.embeddings.arena.data,
                      m_ep.embeddings.shard_from_full(m_dp.embeddings.arena.data),
                      atol=1e-5)
                  and torch.allclose(
                      m_ep.wide_cat.arena.data,
                      m_ep.wide_cat.shard_from_full(m_dp.wide_cat.arena.data),
                      atol=1e-5))
        q.put((rank, bool(ok_dense), bool(ok_emb)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.parametrize("mode,port", [("row", 29727), ("table", 29737)])
def test_ep_window_mode_matches_dp(mode, port):
    for rank, ok_dense, ok_emb in _run(_ep_window_worker, port, extra=(mode,)):
        assert ok_dense, f"rank {rank}: dense params diverged in window mode ({mode})"
        assert ok_emb, f"rank {rank}: embedding shards diverged in window mode ({mode})"


# ===========================================================================
# Table-wise sharding (static all-to-all splits — parallel/ep.py
# TableShardedEmbedding)
# ===========================================================================

def test_assign_features_balanced_deterministic():
    from shifu_amd.parallel.ep import assign_features
    feats = assign_features([1_000_000] * 26, 8)
    sizes = sorted(len(f) for f in feats)
    assert sizes == [3, 3, 3, 3, 3, 3, 4, 4]          # 26 tables over 8 ranks
    assert sorted(j for f in feats for j in f) == list(range(26))
    assert feats == assign_features([1_000_000] * 26, 8)  # deterministic
    # unequal vocabs: greedy keeps loads within the largest table
    feats2 = assign_features([100, 90, 80, 10, 10, 10], 2)
    loads = [sum([100, 90, 80, 10, 10, 10][j] for j in f) for f in feats2]
    assert max(loads) - min(loads) <= 100


def _table_worker(rank, port, q):
    """TableShardedEmbedding parity with the replicated MultiEmbedding:
    init (same RNG stream), forward, and backward (complete per-row grads
    land on the owner)."""
    from shifu_amd.ops.embedding import MultiEmbedding, sparse_rows_values
    from shifu_amd.parallel.ep import TableShardedEmbedding
    try:
        _init(rank, port)
        torch.manual_seed(0)
        vocab = [11, 23, 7]
        ref = MultiEmbedding(vocab, dim=4, seed=7)
        tab = TableShardedEmbedding(vocab, dim=4, seed=7, world=WORLD, rank=rank)

        ok_init = torch.allclose(tab.arena.data,
                                 tab.shard_from_full(ref.arena.data))

        g = torch.Generator().manual_seed(3)
        ids = torch.stack([torch.randint(0, v, (6,), generator=g)
                           for v in vocab], dim=1)
        ok_fwd = torch.allclose(tab(ids), ref(ids), atol=1e-6)

        # backward: different batch per rank; owner must hold the complete
        # mean-over-ranks gradient for its features' rows
        ids_r = (ids + rank) % torch.tensor(vocab)
        out = tab(ids_r)
        (out * (rank + 1.0)).sum().backward()
        grad = tab.arena.grad
        ok_sparse = grad is not None and grad.is_sparse

        ref2 = MultiEmbedding(vocab, dim=4, seed=7)
        for r in range(WORLD):
            o = ref2((ids + r) % torch.tensor(vocab))
            (o * (r + 1.0)).sum().backward()
        expected_full = ref2.arena.grad.coalesce().to_dense() / WORLD
        expected = tab.shard_from_full(expected_full)
        got = torch.zeros_like(expected)
        rows, vals = sparse_rows_values(grad)
        got.index_add_(0, rows, vals.to(got.dtype))
        ok_bwd = torch.allclose(got, expected, atol=1e-5)

        q.put((rank, bool(ok_init), bool(ok_fwd), bool(ok_sparse), bool(ok_bwd)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_table_sharded_parity():
    for rank, ok_init, ok_fwd, ok_sparse, ok_bwd in _run(_table_worker, 29747):
        assert ok_init, f"rank {rank}: table shard init mismatch"
        assert ok_fwd, f"rank {rank}: table forward mismatch"
        assert ok_sparse, f"rank {rank}: table grad not sparse"
        assert ok_bwd, f"rank {rank}: table backward grad mismatch"


def _table_pair_worker(rank, port, q):
    from shifu_amd.parallel.ep import TableShardedEmbedding, table_pair_gather
    try:
        _init(rank, port)
        vocab = [13, 29]
        e1 = TableShardedEmbedding(vocab, dim=1, seed=4, world=WORLD, rank=rank)
        e2 = TableShardedEmbedding(vocab, dim=6, seed=9, world=WORLD, rank=rank)
        g = torch.Generator().manual_seed(8 + rank)
        ids = torch.stack([torch.randint(0, 13, (10,), generator=g),
                           torch.randint(0, 29, (10,), generator=g)], dim=1)
        o1p, o2p = table_pair_gather(e1, e2, ids)
        o1, o2 = e1(ids), e2(ids)
        ok_fwd = (torch.allclose(o1p, o1, atol=1e-6)
                  and torch.allclose(o2p, o2, atol=1e-6))

        (o1p.sum() * 2 + o2p.pow(2).sum()).backward()
        g1p = e1.arena.grad.coalesce().to_dense().clone()
        g2p = e2.arena.grad.coalesce().to_dense().clone()
        e1.arena.grad = None
        e2.arena.grad = None
        o1b, o2b = e1(ids), e2(ids)
        (o1b.sum() * 2 + o2b.pow(2).sum()).backward()
        ok_bwd = (torch.allclose(g1p, e1.arena.grad.coalesce().to_dense(), atol=1e-5)
                  and torch.allclose(g2p, e2.arena.grad.coalesce().to_dense(), atol=1e-5))
        q.put((rank, bool(ok_fwd), bool(ok_bwd)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_table_pair_gather_shared_routing():
    for rank, ok_fwd, ok_bwd in _run(_table_pair_worker, 29749):
        assert ok_fwd, f"rank {rank}: table pair forward mismatch"
        assert ok_bwd, f"rank {rank}: table pair backward mismatch"


def _table_consolidate_worker(rank, port, tmp, q):
    """Trainer export consolidation with table sharding: the chief's exported
    arenas must equal the replicated model's."""
    from shifu_amd.config.run_config import RunConfig
    from shifu_amd.data.csv_loader import TabularDataset
    from shifu_amd.data.synthetic import synthetic_arrays
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.trainer import Trainer
    from shifu_amd.config.model_config import ModelConfig
    import os as _os
    try:
        _init(rank, port)
        vocab = [17, 21]
        dn, ct, tg, w = synthetic_arrays(48, 4, vocab, seed=100 + rank,
                                         weighted=False)
        ds = TabularDataset(dn, ct, tg, w)
        train, valid = ds.split(0.25, seed=1)
        mc = ModelConfig.from_dict({
            "train": {"numTrainEpochs": 1, "validSetRate": 0.2,
                      "params": {"NumHiddenLayers": 1, "NumHiddenNodes": [8],
                                 "ActivationFunc": ["relu"],
                                 "LearningRate": 0.05, "Optimizer": "sgd",
                                 "Loss": "sigmoid_ce", "MiniBatchSize": 16,
                                 "L2Reg": 0.0}}})
        rc = RunConfig(tmp_model_path=_os.path.join(tmp, "ckpt"),
                       final_model_path=_os.path.join(tmp, "final"),
                       device="cpu", batch_size=16)
        model = WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                         sharded_embeddings="table", world=WORLD, rank=rank)
        tr = Trainer(model, mc, rc, train, valid, rank=rank, world_size=WORLD)
        tr.fit()
        dist.barrier()   # the chief's export finishes after the all-gather
        ok = _os.path.exists(_os.path.join(tmp, "final",
                                           "GenericModelConfig.json"))
        q.put((rank, bool(ok)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_table_trainer_export(tmp_path):
    for rank, ok in _run(_table_consolidate_worker, 29751,
                         extra=(str(tmp_path),)):
        assert ok, f"rank {rank}: export artifacts missing"
