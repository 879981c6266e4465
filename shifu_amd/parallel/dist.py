"""Distributed gradient aggregation: RCCL all-reduce over xGMI.

Replaces the reference's SyncReplicasOptimizer parameter-server plane
(gradient push to PS accumulators over gRPC + token-queue barrier,
reference: ssgd_monitor.py:136-142,218-226) with data-parallel all-reduce
(SURVEY.md §2.4 C1):

* dense grads live in ONE flat fp32 buffer (ops/flat.py) split into
  size-`bucket_mb` buckets; each bucket's all-reduce launches as soon as its
  last gradient is accumulated (post-accumulate-grad hooks), overlapping
  communication with the rest of backward.  Bucket size defaults large
  (128 MB): a ring all-reduce over the MI355X's 7x ~153 GB/s point-to-point
  xGMI links is per-link bound, so fewer/larger transfers win.
* embedding grads are SPARSE (rows + values): ranks all-gather their
  coalesced rows over xGMI and each rank applies the merged update locally —
  never a dense all-reduce over a multi-GB arena.
* the backend is torch.distributed "nccl" (RCCL on ROCm) on GPU, "gloo" on
  CPU (multi-process CPU tests).

Local-SGD window mode (SAGN.py:111-167): the trainer calls set_sync(False)
for intra-window steps; gradients then stay local and the window-end step
aggregates the MEAN of the window's gradients (finish() divides the
accumulated gradient by the number of accumulated steps).  Deliberate
divergence from SAGN: the reference also applies a *local* optimizer update
each intra-window step, so its window gradients are evaluated at chained
local points (sagn_monitor.py:122-160); here every window gradient is
evaluated at the window-start point (plain gradient accumulation).  The
global window-end apply uses the window mean in both.

Quorum / partial participation (REPLICAS_TO_AGGREGATE_RATIO, reference
ssgd.py:19; SAGN.py:161 uses 90%): with quorum_ratio q < 1, each sync step
aggregates gradients from only ceil(q*world) ranks — the contributor set
rotates deterministically so every rank participates equally over time, and
the update divides by the contributor count.  The reference's PS applies
whichever q*N gradients ARRIVE first (dropping stragglers); a lockstep
all-reduce has no arrival order, so the rotation reproduces the gradient
statistics (each update is a q-fraction sample mean) rather than the
straggler selection — on one xGMI-coupled node there are no slow hosts to
drop.  Non-contributing ranks join the same collectives with zeroed
gradients, so the parameter trajectory stays identical on every rank.
"""
from __future__ import annotations

import datetime
import math
import os
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from shifu_amd.ops.flat import FlatParams


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def init_distributed(backend: str = "auto", master_addr: str = "127.0.0.1",
                     master_port: int = 29511, timeout_s: int = 600) -> Tuple[int, int, torch.device]:
    """Init from torchrun-style env (RANK/WORLD_SIZE/LOCAL_RANK) or single-process.

    Returns (rank, world_size, device)."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
        device = torch.device("cuda", local_rank % max(torch.cuda.device_count(), 1))
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", master_addr)
        os.environ.setdefault("MASTER_PORT", str(master_port))
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=timeout_s))
    return rank, world, device


def destroy_distributed() -> None:
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()


class GradAggregator:
    """Bucketed, backward-overlapped all-reduce on a FlatParams arena +
    sparse allgather for embedding arenas."""

    def __init__(self, flat: FlatParams, emb_params: Optional[List[torch.nn.Parameter]] = None,
                 bucket_mb: int = 128, overlap: bool = True,
                 quorum_ratio: float = 1.0):
        self.flat = flat
        self.emb_params = list(emb_params or [])
        self.bucket_bytes = int(bucket_mb) * (1 << 20)
        self.sync_enabled = True
        self.overlap = overlap and is_distributed()
        self.quorum_ratio = float(quorum_ratio)
        if not (0.0 < self.quorum_ratio <= 1.0):
            raise ValueError(f"quorum_ratio must be in (0, 1], got {quorum_ratio}")
        if self.quorum_ratio < 1.0 and any(
                getattr(p, "_is_ep_sharded", False) for p in self.emb_params):
            raise ValueError(
                "quorum_ratio < 1 requires replicated embeddings (emb_mode=dp): "
                "EP-sharded arenas receive their complete row gradients through "
                "the all-to-all, outside the quorum aggregation plane")
        self._accum_steps = 0          # local steps since the last finish()
        self._sync_count = 0           # finished sync steps (quorum rotation)
        self._works: List[dist.Work] = []
        self._buckets: List[Tuple[int, int]] = []          # (start, numel) in flat_grad
        self._param_bucket: dict = {}                      # param -> bucket index
        self._pending: List[int] = []                      # params left per bucket
        self._hook_handles = []
        if is_distributed() and flat.numel():
            self._build_buckets()
            if self.overlap:
                self._register_hooks()

    # ---------------------------------------------------------------- buckets
    def _build_buckets(self) -> None:
        """Group consecutive params (in REVERSE order — backward fills the last
        layer's grads first) into ~bucket_bytes buckets of the flat buffer."""
        elt = self.flat.flat_grad.element_size()
        per_bucket = max(self.bucket_bytes // elt, 1)
        cur: List[int] = []
        cur_n = 0
        buckets_params: List[List[int]] = []
        order = list(range(len(self.flat.params)))[::-1]
        for pi in order:
            _, n = self.flat._offsets[pi]
            cur.append(pi)
            cur_n += n
            if cur_n >= per_bucket:
                buckets_params.append(cur)
                cur, cur_n = [], 0
        if cur:
            buckets_params.append(cur)
        self._buckets = []
        self._bucket_params = buckets_params
        for bi, plist in enumerate(buckets_params):
            start = min(self.flat._offsets[pi][0] for pi in plist)
            end = max(self.flat._offsets[pi][0] + self.flat._offsets[pi][1] for pi in plist)
            self._buckets.append((start, end - start))
            for pi in plist:
                self._param_bucket[self.flat.params[pi]] = bi
        self._param_offset = {p: off for p, off in zip(self.flat.params, self.flat._offsets)}
        self._pending = [len(p) for p in self._bucket_params]

    def _register_hooks(self) -> None:
        for p in self.flat.params:
            h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hook_handles.append(h)

    def _on_grad_ready(self, p: torch.nn.Parameter) -> None:
        if not self.sync_enabled or not is_distributed():
            return
        # repair out-of-place .grad into the flat view before reducing
        off, n = self._param_offset[p]
        view = self.flat.flat_grad[off:off + n]
        if p.grad is not None and p.grad.data_ptr() != view.data_ptr():
            view.copy_(p.grad.reshape(-1))
            p.grad = view.view_as(p.data)
        bi = self._param_bucket[p]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            start, numel = self._buckets[bi]
            chunk = self.flat.flat_grad[start:start + numel]
            if not self._contributing():
                chunk.zero_()  # quorum: join the collective with zeros
            self._works.append(dist.all_reduce(chunk, op=dist.ReduceOp.SUM, async_op=True))

    # ------------------------------------------------------------------- api
    def set_sync(self, enabled: bool) -> None:
        """Called once per local step, BEFORE backward.  Counts window steps
        so finish() can apply the window-mean scaling."""
        self.sync_enabled = enabled
        self._accum_steps += 1

    def _n_contributors(self) -> int:
        world = dist.get_world_size() if is_distributed() else 1
        return max(1, math.ceil(self.quorum_ratio * world))

    def _contributing(self, rank: Optional[int] = None) -> bool:
        """Deterministic rotating contributor set for quorum < 1: every rank
        computes the same schedule, so collectives stay matched."""
        if self.quorum_ratio >= 1.0 or not is_distributed():
            return True
        world = dist.get_world_size()
        r = dist.get_rank() if rank is None else rank
        return (r + self._sync_count) % world < self._n_contributors()

    def finish(self) -> None:
        """Call after backward(): waits for in-flight buckets / runs the
        non-overlapped path, averages (over contributors x window steps), and
        aggregates sparse embedding grads."""
        world = dist.get_world_size() if is_distributed() else 1
        n_contrib = self._n_contributors()
        steps = max(self._accum_steps, 1)
        if self.flat.numel() and self.flat.flat.is_cuda:
            from shifu_amd.ops.linear import drain_wgrad_events
            drain_wgrad_events()
        if is_distributed() and self.sync_enabled and self.flat.numel():
            if self.overlap:
                for w in self._works:
                    w.wait()
                self._works.clear()
                # any bucket whose hook chain never completed this step (a
                # param without grad, or hooks skipped under window mode) is
                # reduced now so every rank joins the same collectives
                for bi, left in enumerate(self._pending):
                    if left > 0:
                        start, numel = self._buckets[bi]
                        chunk = self.flat.flat_grad[start:start + numel]
                        if not self._contributing():
                            chunk.zero_()
                        dist.all_reduce(chunk, op=dist.ReduceOp.SUM)
            else:
                self.flat.sync_grads()
                if not self._contributing():
                    self.flat.flat_grad.zero_()
                dist.all_reduce(self.flat.flat_grad, op=dist.ReduceOp.SUM)
            self.flat.flat_grad.div_(n_contrib * steps)
        elif self.sync_enabled and self.flat.numel() and steps > 1:
            # single-process window mode: same window-mean scaling
            self.flat.sync_grads()
            self.flat.flat_grad.div_(steps)
        self._pending = [len(p) for p in getattr(self, "_bucket_params", [])]

        if self.sync_enabled:
            for p in self.emb_params:
                if getattr(p, "_is_ep_sharded", False):
                    # sharded arenas: backward all-to-all already delivered the
                    # complete per-row gradient (mean over ranks); window mean
                    # still applies locally
                    if steps > 1 and p.grad is not None:
                        p.grad = p.grad * (1.0 / steps)
                    continue
                if is_distributed():
                    self._aggregate_sparse(p, n_contrib * steps)
                elif steps > 1 and p.grad is not None:
                    p.grad = p.grad * (1.0 / steps)
            self._accum_steps = 0
            self._sync_count += 1

    def _aggregate_sparse(self, p: torch.nn.Parameter, divisor: int) -> None:
        """All-gather sparse rows+values and rebuild the merged grad; `divisor`
        is contributors x window steps (the mean denominator)."""
        if p.grad is None:
            return
        if not p.grad.is_sparse:  # dense emb grad (small test arenas)
            if not self._contributing():
                p.grad.zero_()
            dist.all_reduce(p.grad, op=dist.ReduceOp.SUM)
            p.grad.div_(divisor)
            return
        from shifu_amd.ops.embedding import sparse_rows_values
        world = dist.get_world_size()
        idx, vals = sparse_rows_values(p.grad)
        idx, vals = idx.contiguous(), vals.contiguous()
        if not self._contributing():
            vals = torch.zeros_like(vals)
        dev = vals.device
        n = torch.tensor([idx.numel()], device=dev, dtype=torch.int64)
        ns = [torch.zeros_like(n) for _ in range(world)]
        dist.all_gather(ns, n)
        nmax = int(torch.stack(ns).max())
        if nmax == 0:
            p.grad = None
            return
        idx_pad = torch.zeros(nmax, device=dev, dtype=idx.dtype)
        val_pad = torch.zeros(nmax, vals.shape[1], device=dev, dtype=vals.dtype)
        idx_pad[:idx.numel()] = idx
        val_pad[:idx.numel()] = vals
        idx_out = [torch.zeros_like(idx_pad) for _ in range(world)]
        val_out = [torch.zeros_like(val_pad) for _ in range(world)]
        dist.all_gather(idx_out, idx_pad)
        dist.all_gather(val_out, val_pad)
        all_idx = torch.cat([t[:int(c)] for t, c in zip(idx_out, ns)])
        all_val = (torch.cat([t[:int(c)] for t, c in zip(val_out, ns)]) / divisor).to(p.dtype)
        p.grad = torch.sparse_coo_tensor(all_idx.unsqueeze(0), all_val, p.shape)

    def remove_hooks(self) -> None:
        for h in self._hook_handles:
            h.remove()
        self._hook_handles.clear()
