"""Rank-0 checkpoint / resume.

Successor of the reference's MonitoredTrainingSession autosave/restore to
TMP_MODEL_PATH (reference: ssgd_monitor.py:251-257): rank 0 writes
(model state, optimizer moments, epoch, global step, RNG states) at epoch
cadence; on restart every rank loads the newest checkpoint and training
resumes from the next epoch (the resume-aware progress offset of
AMRMCallbackHandler.getProgress:224-244).
"""
from __future__ import annotations

import os
import re
from typing import Optional

import torch


CKPT_RE = re.compile(r"ckpt-(\d+)\.pt$")


def checkpoint_path(ckpt_dir: str, epoch: int) -> str:
    return os.path.join(ckpt_dir, f"ckpt-{epoch}.pt")


def latest_checkpoint(ckpt_dir: str) -> Optional[str]:
    if not os.path.isdir(ckpt_dir):
        return None
    best, best_epoch = None, -1
    for name in os.listdir(ckpt_dir):
        m = CKPT_RE.search(name)
        if m and int(m.group(1)) > best_epoch:
            best_epoch = int(m.group(1))
            best = os.path.join(ckpt_dir, name)
    return best


def save_checkpoint(ckpt_dir: str, epoch: int, global_step: int,
                    model: torch.nn.Module, optimizer, keep_last: int = 3,
                    extra: Optional[dict] = None) -> str:
    os.makedirs(ckpt_dir, exist_ok=True)
    path = checkpoint_path(ckpt_dir, epoch)
    tmp = path + ".tmp"
    torch.save({
        "epoch": epoch,
        "global_step": global_step,
        "model": {k: v.cpu() for k, v in model.state_dict().items()},
        "optimizer": _optim_state_cpu(optimizer),
        "torch_rng": torch.get_rng_state(),
        "extra": extra or {},
    }, tmp)
    os.replace(tmp, path)  # atomic publish — a crashed writer never corrupts
    # prune old checkpoints
    ckpts = sorted(
        (int(CKPT_RE.search(n).group(1)), n) for n in os.listdir(ckpt_dir) if CKPT_RE.search(n))
    for _, name in ckpts[:-keep_last]:
        try:
            os.remove(os.path.join(ckpt_dir, name))
        except OSError:
            pass
    return path


def _optim_state_cpu(optimizer) -> dict:
    sd = optimizer.state_dict()
    out = {"step_count": sd["step_count"]}
    out["m"] = sd["m"].cpu() if sd.get("m") is not None else None
    out["v"] = sd["v"].cpu() if sd.get("v") is not None else None
    out["emb_state"] = {k: t.cpu() for k, t in (sd.get("emb_state") or {}).items()}
    return out


def load_checkpoint(path: str, model: torch.nn.Module, optimizer=None,
                    device: Optional[torch.device] = None) -> dict:
    blob = torch.load(path, map_location="cpu", weights_only=False)
    state = blob["model"]
    if device is not None:
        state = {k: v.to(device) for k, v in state.items()}
    model.load_state_dict(state)
    if optimizer is not None and blob.get("optimizer") is not None:
        opt_sd = blob["optimizer"]
        if device is not None:
            opt_sd = {k: (v.to(device) if torch.is_tensor(v) else
                          ({kk: tt.to(device) for kk, tt in v.items()} if isinstance(v, dict) else v))
                      for k, v in opt_sd.items()}
        optimizer.load_state_dict(opt_sd)
    return {"epoch": blob["epoch"], "global_step": blob["global_step"],
            "extra": blob.get("extra", {})}
