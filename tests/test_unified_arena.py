"""Unified [R, D+2] wide+deep arena (ROADMAP item 3): gather/grad
correctness, end-to-end training, export-time logical split, and EP parity."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from shifu_amd.config.model_config import ModelConfig
from shifu_amd.config.run_config import RunConfig
from shifu_amd.data.csv_loader import TabularDataset
from shifu_amd.data.synthetic import synthetic_arrays
from shifu_amd.models.wide_deep import WideDeep
from shifu_amd.models.deepfm import DeepFM
from shifu_amd.ops.embedding import UnifiedMultiEmbedding, unified_col_scale
from shifu_amd.train.trainer import Trainer

WORLD = 2


def _mc(epochs=3, lr=0.02):
    return ModelConfig.from_dict({
        "train": {"numTrainEpochs": epochs, "validSetRate": 0.2,
                  "params": {"NumHiddenLayers": 1, "NumHiddenNodes": [16],
                             "ActivationFunc": ["relu"], "LearningRate": lr,
                             "Optimizer": "adam", "Loss": "sigmoid_ce",
                             "MiniBatchSize": 64, "L2Reg": 0.0}}})


def test_unified_gather_split_matches_reference():
    emb = UnifiedMultiEmbedding([9, 14], dim=8, seed=3)
    g = torch.Generator().manual_seed(1)
    ids = torch.stack([torch.randint(0, 9, (5,), generator=g),
                       torch.randint(0, 14, (5,), generator=g)], dim=1)
    dense = torch.randn(5, 3, generator=g)
    out, wide = emb.gather_split(ids, dense)
    assert out.shape == (5, 3 + 2 * 8) and wide.shape == (5, 2)

    flat = emb.flat_ids(ids).reshape(-1)
    ref = emb.arena.detach().index_select(0, flat)         # [10, 10]
    assert torch.allclose(out[:, 3:], ref[:, :8].reshape(5, 16))
    assert torch.allclose(wide, ref[:, 8].reshape(5, 2))
    assert torch.allclose(out[:, :3], dense)
    # pad column must be zero-initialized
    assert torch.all(emb.arena.data[:, 9] == 0)


def test_unified_backward_packs_wide_and_deep():
    emb = UnifiedMultiEmbedding([7], dim=4, seed=0)
    ids = torch.tensor([[2], [5], [2]])
    dense = torch.randn(3, 2)
    out, wide = emb.gather_split(ids, dense)
    loss = (out[:, 2:] * 2.0).sum() + (wide * 3.0).sum()
    loss.backward()
    g = emb.arena.grad
    assert g is not None and g.is_sparse
    gd = g.coalesce().to_dense()
    # deep cols: rows 2 (twice) and 5 get 2.0 per element
    assert torch.allclose(gd[2, :4], torch.full((4,), 4.0))
    assert torch.allclose(gd[5, :4], torch.full((4,), 2.0))
    # wide col: 3.0 per occurrence
    assert float(gd[2, 4]) == 6.0 and float(gd[5, 4]) == 3.0
    # pad col gets no gradient
    assert torch.all(gd[:, 5] == 0)


def _train(model, tmp, n=1500, epochs=4):
    dn, ct, tg, w = synthetic_arrays(n, 6, (40, 60), seed=11, weighted=False)
    ds = TabularDataset(dn, ct, tg, w)
    train, valid = ds.split(0.2, seed=1)
    rc = RunConfig(tmp_model_path=os.path.join(tmp, "ckpt"),
                   final_model_path=os.path.join(tmp, "final"),
                   device="cpu", batch_size=64)
    tr = Trainer(model, _mc(epochs=epochs), rc, train, valid)
    tr.fit()
    return tr


def test_unified_wide_deep_learns(tmp_path):
    m = WideDeep(6, [40, 60], 8, [16], ["relu"], seed=2, unified=True)
    tr = _train(m, str(tmp_path), epochs=6)
    ev = tr.evaluate(tr.valid_data)
    # this tiny config tops out near 0.58 for BOTH layouts; the split-vs-
    # unified comparison below is the real quality gate
    assert ev["auc"] > 0.55, f"unified Wide&Deep failed to learn: {ev}"


def test_unified_deepfm_learns(tmp_path):
    m = DeepFM(6, [40, 60], 8, [16], ["relu"], seed=2, unified=True)
    tr = _train(m, str(tmp_path), epochs=6)
    ev = tr.evaluate(tr.valid_data)
    assert ev["auc"] > 0.55, f"unified DeepFM failed to learn: {ev}"


def test_unified_export_splits_to_legacy_layout(tmp_path):
    """Exported unified models reload as the split layout and score
    IDENTICALLY (same math, exact weights)."""
    from shifu_amd.train.export import load_exported
    m = WideDeep(6, [40, 60], 8, [16], ["relu"], seed=2, unified=True)
    tr = _train(m, str(tmp_path), epochs=2)
    loaded = load_exported(os.path.join(str(tmp_path), "final"))
    assert not loaded.unified and hasattr(loaded, "wide_cat")
    assert loaded.wide_cat.arena.shape == (100, 1)
    assert loaded.embeddings.arena.shape == (100, 8)

    g = torch.Generator().manual_seed(5)
    dense = torch.randn(16, 6, generator=g)
    cats = torch.stack([torch.randint(0, 40, (16,), generator=g),
                        torch.randint(0, 60, (16,), generator=g)], dim=1)
    with torch.no_grad():
        a = m(dense, cats)
        b = loaded(dense, cats)
    assert torch.allclose(a, b, atol=1e-5), "unified vs exported-split scores differ"


def test_unified_auc_parity_with_split(tmp_path):
    """The adagrad-normalization semantics change (wide weight normalized by
    the row's combined grad) must not hurt model quality: unified AUC within
    0.02 of the split layout on the same data."""
    m_s = WideDeep(6, [40, 60], 8, [16], ["relu"], seed=2, unified=False)
    tr_u = _train(WideDeep(6, [40, 60], 8, [16], ["relu"], seed=2, unified=True),
                  str(tmp_path / "u"))
    tr_s = _train(m_s, str(tmp_path / "s"))
    auc_u = tr_u.evaluate(tr_u.valid_data)["auc"]
    auc_s = tr_s.evaluate(tr_s.valid_data)["auc"]
    assert auc_u > auc_s - 0.02, f"unified AUC {auc_u} vs split {auc_s}"


# ------------------------------------------------------------------ EP parity
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


def _unified_ep_worker(rank, port, q):
    """Unified EP (table-sharded [R, D+2]) forward == replicated unified;
    one training step keeps the shard equal to its slice of the DP twin."""
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.ops.optim import FusedOptimizer
    from shifu_amd.parallel.dist import GradAggregator
    try:
        _init(rank, port)
        torch.manual_seed(0)
        vocab = [23, 31]

        def build(sharded):
            return WideDeep(4, vocab, 4, [8], ["relu"], seed=5,
                            sharded_embeddings=sharded, world=WORLD,
                            rank=rank, unified=True)

        # forward parity on an identical batch
        m_ep, m_dp = build("table"), build(False)
        g = torch.Generator().manual_seed(7)
        dense = torch.randn(6, 4, generator=g)
        cats = torch.stack([torch.randint(0, 23, (6,), generator=g),
                            torch.randint(0, 31, (6,), generator=g)], dim=1)
        ok_fwd = torch.allclose(m_ep(dense, cats), m_dp(dense, cats), atol=1e-5)

        def one_step(model):
            dp, ep = split_params(model)
            flat = FlatParams(dp)
            agg = GradAggregator(flat, ep, bucket_mb=1)
            opt = FusedOptimizer(flat, ep, optimizer="sgd", lr=0.1,
                                 l2_reg=0.0, emb_optimizer="sgd", emb_lr=0.1)
            g2 = torch.Generator().manual_seed(100 + rank)
            d = torch.randn(8, 4, generator=g2)
            c = torch.stack([torch.randint(0, 23, (8,), generator=g2),
                             torch.randint(0, 31, (8,), generator=g2)], dim=1)
            y = (torch.rand(8, generator=g2) > 0.5).float()
            loss = weighted_loss(model(d, c), y, torch.ones(8), "sigmoid_ce")
            loss.backward()
            agg.finish()
            opt.step()
            return model

        m_ep = one_step(build("table"))
        m_dp = one_step(build(False))
        # EP arenas are [.., D+2]; the replicated unified arena is [.., D+4]
        # (in-row adagrad accumulator): compare the deep+wide columns
        D = m_ep.embed_dim
        ok_step = torch.allclose(
            m_ep.embeddings.arena.data[:, :D + 1],
            m_ep.embeddings.shard_from_full(
                m_dp.embeddings.arena.data[:, :D + 2])[:, :D + 1],
            atol=1e-5)
        q.put((rank, bool(ok_fwd), bool(ok_step)))
        dist.barrier()
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_unified_ep_matches_replicated():
    for rank, ok_fwd, ok_step in _run(_unified_ep_worker, 29753):
        assert ok_fwd, f"rank {rank}: unified EP forward mismatch"
        assert ok_step, f"rank {rank}: unified EP training-step mismatch"
