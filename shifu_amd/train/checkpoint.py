"""Checkpoint / resume (EP-shard aware).

Successor of the reference's MonitoredTrainingSession autosave/restore to
TMP_MODEL_PATH (reference: ssgd_monitor.py:251-257): at epoch cadence the
chief writes (dense model state, optimizer moments, epoch, global step, RNG
states); on restart every rank loads the newest COMPLETE checkpoint and
training resumes from the next epoch (the resume-aware progress offset of
AMRMCallbackHandler.getProgress:224-244).

Expert-parallel sharded embeddings (parallel/ep.py) hold DIFFERENT rows on
every rank, so a chief-only checkpoint cannot represent them.  Layout:

* ``ckpt-<E>.pt``              — rank 0: everything EXCEPT sharded arenas
                                 (records the sharded names + world size)
* ``ckpt-<E>.shard<R>of<W>.pt``— every rank: its arena rows + that arena's
                                 rowwise optimizer state

A checkpoint epoch is COMPLETE only when the main file and (if sharded) all
W shard files exist — ranks pick the newest complete epoch independently but
deterministically (shared filesystem, single node).  Resuming a sharded
model from an un-sharded (replicated) checkpoint re-shards the full arenas
row%world; world-size changes across restarts of a sharded run hard-fail.
"""
from __future__ import annotations

import os
import re
from typing import Dict, List, Optional

import torch


CKPT_RE = re.compile(r"ckpt-(-?\d+)\.pt$")
SHARD_RE = re.compile(r"ckpt-(-?\d+)\.shard(\d+)of(\d+)\.pt$")


def checkpoint_path(ckpt_dir: str, epoch: int) -> str:
    return os.path.join(ckpt_dir, f"ckpt-{epoch}.pt")


def shard_checkpoint_path(ckpt_dir: str, epoch: int, rank: int, world: int) -> str:
    return os.path.join(ckpt_dir, f"ckpt-{epoch}.shard{rank}of{world}.pt")


def sharded_param_names(model: torch.nn.Module) -> List[str]:
    return [n for n, p in model.named_parameters()
            if getattr(p, "_is_ep_sharded", False)]


def _shard_epochs(ckpt_dir: str) -> Dict[int, set]:
    """epoch -> {(rank, world)} of shard files present."""
    out: Dict[int, set] = {}
    for name in os.listdir(ckpt_dir):
        m = SHARD_RE.search(name)
        if m:
            out.setdefault(int(m.group(1)), set()).add(
                (int(m.group(2)), int(m.group(3))))
    return out


def latest_checkpoint(ckpt_dir: str, world: int = 1) -> Optional[str]:
    """Newest COMPLETE checkpoint: the main file plus, when that epoch has
    shard files, the full shard set for SOME world size (load validates the
    size against the caller's).  Deterministic across ranks."""
    if not os.path.isdir(ckpt_dir):
        return None
    shard_sets = _shard_epochs(ckpt_dir)
    epochs = sorted((int(CKPT_RE.search(n).group(1)) for n in os.listdir(ckpt_dir)
                     if CKPT_RE.search(n)), reverse=True)
    for e in epochs:
        shards = shard_sets.get(e)
        if shards is None:
            return checkpoint_path(ckpt_dir, e)   # un-sharded checkpoint
        worlds = {w for _, w in shards}
        if any(all((r, w) in shards for r in range(w)) for w in worlds):
            return checkpoint_path(ckpt_dir, e)
    return None


def _atomic_save(blob: dict, path: str) -> None:
    tmp = path + ".tmp"
    torch.save(blob, tmp)
    os.replace(tmp, path)  # atomic publish — a crashed writer never corrupts


def save_checkpoint(ckpt_dir: str, epoch: int, global_step: int,
                    model: torch.nn.Module, optimizer, keep_last: int = 3,
                    extra: Optional[dict] = None, rank: int = 0,
                    world: int = 1, stamp: Optional[int] = None) -> str:
    """Write this rank's part of checkpoint `epoch`.  Rank 0 writes the main
    file; every rank owning EP shards writes its shard file.  Call from ALL
    ranks when the model holds ShardedEmbeddings."""
    os.makedirs(ckpt_dir, exist_ok=True)
    path = checkpoint_path(ckpt_dir, epoch)
    sharded = sharded_param_names(model)
    shard_idx = set()
    if optimizer is not None:
        shard_idx = {i for i, p in enumerate(optimizer.emb_params)
                     if getattr(p, "_is_ep_sharded", False)}

    if sharded:
        params = dict(model.named_parameters())
        arena_state = {}
        if optimizer is not None:
            by_param = {id(p): i for i, p in enumerate(optimizer.emb_params)}
            for n in sharded:
                i = by_param.get(id(params[n]))
                acc = (optimizer.state_dict().get("emb_state") or {}).get(i)
                if acc is not None:
                    arena_state[n] = acc.cpu()
        _atomic_save({
            "epoch": epoch, "rank": rank, "world": world, "stamp": stamp,
            "arenas": {n: params[n].data.cpu() for n in sharded},
            "arena_state": arena_state,
        }, shard_checkpoint_path(ckpt_dir, epoch, rank, world))

    if rank == 0:
        skip = set(sharded)
        _atomic_save({
            "epoch": epoch,
            "stamp": stamp,
            "global_step": global_step,
            "model": {k: v.cpu() for k, v in model.state_dict().items()
                      if k not in skip},
            "optimizer": _optim_state_cpu(optimizer, skip_emb=shard_idx),
            "sharded": ({"names": sorted(sharded), "world": world}
                        if sharded else None),
            "torch_rng": torch.get_rng_state(),
            "extra": extra or {},
        }, path)

    _prune(ckpt_dir, keep_last, rank, world, prune_main=(rank == 0))
    return path


def _prune(ckpt_dir: str, keep_last: int, rank: int, world: int,
           prune_main: bool) -> None:
    """Each rank prunes only the files it writes (its shards; rank 0 also the
    main files), keyed by epoch."""
    mains = sorted(int(CKPT_RE.search(n).group(1)) for n in os.listdir(ckpt_dir)
                   if CKPT_RE.search(n))
    drop = set(mains[:-keep_last])
    for e in drop:
        victims = []
        if prune_main:
            victims.append(checkpoint_path(ckpt_dir, e))
        victims.append(shard_checkpoint_path(ckpt_dir, e, rank, world))
        for v in victims:
            try:
                os.remove(v)
            except OSError:
                pass


def _optim_state_cpu(optimizer, skip_emb=frozenset()) -> Optional[dict]:
    if optimizer is None:
        return None
    sd = optimizer.state_dict()
    out = {"step_count": sd["step_count"]}
    out["m"] = sd["m"].cpu() if sd.get("m") is not None else None
    out["v"] = sd["v"].cpu() if sd.get("v") is not None else None
    out["emb_state"] = {k: t.cpu() for k, t in (sd.get("emb_state") or {}).items()
                        if k not in skip_emb}
    return out


def load_checkpoint(path: str, model: torch.nn.Module, optimizer=None,
                    device: Optional[torch.device] = None, rank: int = 0,
                    world: int = 1) -> dict:
    blob = torch.load(path, map_location="cpu", weights_only=False)
    sharded_here = sharded_param_names(model)
    info = blob.get("sharded")
    state = dict(blob["model"])

    if sharded_here:
        params = dict(model.named_parameters())
        if info:  # sharded checkpoint -> shard files must match this topology
            if int(info["world"]) != world:
                raise RuntimeError(
                    f"EP checkpoint {path} was written at world={info['world']} "
                    f"but this run has world={world}; re-shard is not supported "
                    "across world sizes — export + restart, or keep the size")
            if sorted(info["names"]) != sorted(sharded_here):
                raise RuntimeError(
                    f"EP checkpoint arenas {info['names']} do not match the "
                    f"model's sharded params {sorted(sharded_here)}")
            spath = shard_checkpoint_path(os.path.dirname(path),
                                          int(blob["epoch"]), rank, world)
            sblob = torch.load(spath, map_location="cpu", weights_only=False)
            if sblob.get("stamp") != blob.get("stamp"):
                raise RuntimeError(
                    f"checkpoint {path} and shard {spath} come from different "
                    "mid-epoch save points (the job died during a cadence "
                    "save); delete this epoch's ckpt files to resume from "
                    "the previous complete checkpoint")
            for n in sharded_here:
                src = sblob["arenas"][n]
                if src.shape != params[n].shape:
                    raise RuntimeError(
                        f"shard {spath} arena {n} has shape {tuple(src.shape)}, "
                        f"model expects {tuple(params[n].shape)}")
                params[n].data.copy_(src.to(params[n].device, params[n].dtype))
            _load_arena_state(model, optimizer, sblob.get("arena_state") or {},
                              sharded_here)
        else:     # replicated checkpoint -> re-shard via the owning module's
                  # topology (row%world or by-feature ranges)
            for n in sharded_here:
                full = state.pop(n, None)
                if full is None:
                    raise RuntimeError(
                        f"checkpoint {path} lacks arena {n} needed by the "
                        "EP-sharded model")
                shard = _owner_module(model, n).shard_from_full(full)
                if shard.shape != params[n].shape:
                    raise RuntimeError(
                        f"re-shard of {n}: {tuple(full.shape)} -> "
                        f"{tuple(shard.shape)} != model "
                        f"{tuple(params[n].shape)}")
                params[n].data.copy_(shard.to(params[n].device, params[n].dtype))
            _reshard_emb_state(model, optimizer, blob, sharded_here)
    elif info:
        raise RuntimeError(
            f"checkpoint {path} holds EP shards (world={info['world']}) but the "
            "model has no sharded embeddings; consolidate via export or resume "
            "with the same emb_mode")

    for n in sharded_here:
        state.pop(n, None)
    # older checkpoints persisted embedding topology buffers (offsets/sizes/
    # perm/...) — these are derived from vocab_sizes and never loaded
    _TOPO = {"offsets", "sizes", "perm", "inv_perm", "recv_offsets"}
    expected = set(model.state_dict().keys())
    state = {k: v for k, v in state.items()
             if not (k not in expected and k.rsplit(".", 1)[-1] in _TOPO)}
    if device is not None:
        state = {k: v.to(device) for k, v in state.items()}
    missing, unexpected = model.load_state_dict(state, strict=False)
    missing = [m for m in missing if m not in sharded_here]
    if missing or unexpected:
        raise RuntimeError(f"checkpoint/model mismatch: missing={missing} "
                           f"unexpected={unexpected}")

    if optimizer is not None and blob.get("optimizer") is not None:
        opt_sd = dict(blob["optimizer"])
        if sharded_here:
            # sharded arenas' rowwise state was loaded above (shard file /
            # re-shard); keep load_state_dict away from those indices
            shard_idx = {i for i, p in enumerate(optimizer.emb_params)
                         if getattr(p, "_is_ep_sharded", False)}
            opt_sd["emb_state"] = {k: v for k, v in
                                   (opt_sd.get("emb_state") or {}).items()
                                   if int(k) not in shard_idx}
        if device is not None:
            opt_sd = {k: (v.to(device) if torch.is_tensor(v) else
                          ({kk: tt.to(device) for kk, tt in v.items()} if isinstance(v, dict) else v))
                      for k, v in opt_sd.items()}
        optimizer.load_state_dict(opt_sd)
    return {"epoch": blob["epoch"], "global_step": blob["global_step"],
            "extra": blob.get("extra", {})}


def _owner_module(model: torch.nn.Module, pname: str) -> torch.nn.Module:
    mod = model
    for part in pname.split(".")[:-1]:
        mod = getattr(mod, part)
    return mod


def _emb_index_of(model, optimizer, name: str) -> Optional[int]:
    p = dict(model.named_parameters())[name]
    for i, q in enumerate(optimizer.emb_params):
        if q is p:
            return i
    return None


def _load_arena_state(model, optimizer, arena_state: dict, names) -> None:
    if optimizer is None:
        return
    for n in names:
        acc = arena_state.get(n)
        if acc is None:
            continue
        i = _emb_index_of(model, optimizer, n)
        if i is not None and i in optimizer.emb_state:
            optimizer.emb_state[i].copy_(acc.to(optimizer.emb_state[i].device))


def _reshard_emb_state(model, optimizer, blob, names) -> None:
    """Replicated checkpoint -> EP model: rowwise accumulators shard the same
    row pattern as their arenas (owning module's shard_from_full).  Index
    mapping relies on split_params ordering being identical across the
    save/load model builds."""
    if optimizer is None:
        return
    full_state = (blob.get("optimizer") or {}).get("emb_state") or {}
    for n in names:
        i = _emb_index_of(model, optimizer, n)
        if i is None or i not in optimizer.emb_state:
            continue
        acc = full_state.get(i)
        if acc is not None:
            shard = _owner_module(model, n).shard_from_full(acc)
            if shard.shape == optimizer.emb_state[i].shape:
                optimizer.emb_state[i].copy_(
                    shard.to(optimizer.emb_state[i].device))
