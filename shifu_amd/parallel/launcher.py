"""Single-node multi-GPU launcher with supervision.

Collapses the reference's L1-L4 Java stack (client -> YARN AM -> session FSM
-> per-container executor) into one process that:

* forks one rank process per GPU (torch.multiprocessing), passing the
  torchrun-style env (RANK/WORLD_SIZE/LOCAL_RANK/MASTER_*) for RCCL
  rendezvous — no ZooKeeper, no embedded cluster spec;
* receives per-epoch TrainingIntermediateResults and heartbeats over an
  in-process queue (replaces Python->TCP->Java->ZK->AM, §2.4 C5), aggregates
  them per epoch exactly like TensorflowSession.doStatistic:515-549, and
  appends to a console board file;
* supervises: a dead or heartbeat-silent rank aborts the step and the whole
  job restarts from the newest checkpoint (up to max_rank_restarts times) —
  the single-node successor of the backup-worker wake-up machinery
  (SURVEY.md §5.3); rank-0 failure semantics match chief failure
  (TensorflowSession.java:443-450).
"""
from __future__ import annotations

import os
import queue
import sys
import time
import traceback
from collections import defaultdict
from typing import Callable, Dict, List, Optional

import torch
import torch.multiprocessing as mp

from shifu_amd.config.model_config import ModelConfig
from shifu_amd.config.run_config import RunConfig
from shifu_amd.train.metrics import (ConsoleBoard, EpochStats,
                                     TrainingIntermediateResult)

MSG_METRIC, MSG_HEARTBEAT, MSG_DONE, MSG_ERROR = "metric", "hb", "done", "error"


def _rank_main(rank: int, world: int, rc_dict: dict, mc_dict: dict,
               entry: Callable, q) -> None:
    """Child process body: set env, run the user entry(rank, world, rc, mc, sink)."""
    # per-rank log files (successor of YARN container logs)
    log_dir = rc_dict.get("log_dir") or "./logs"
    try:
        os.makedirs(log_dir, exist_ok=True)
        logf = open(os.path.join(log_dir, f"rank-{rank}.log"), "a", buffering=1)
        sys.stdout = logf
        sys.stderr = logf
    except OSError:
        pass
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ.setdefault("MASTER_ADDR", rc_dict.get("master_addr", "127.0.0.1"))
    os.environ.setdefault("MASTER_PORT", str(rc_dict.get("master_port", 29511)))
    rc = RunConfig(**{k: v for k, v in rc_dict.items() if k in RunConfig.__dataclass_fields__})
    mc = ModelConfig.from_dict(mc_dict) if mc_dict else ModelConfig()

    def sink(r: TrainingIntermediateResult):
        q.put((MSG_METRIC, rank, r))

    def heartbeat():
        q.put((MSG_HEARTBEAT, rank, time.time()))

    try:
        entry(rank, world, rc, mc, sink, heartbeat)
        q.put((MSG_DONE, rank, None))
    except Exception:
        q.put((MSG_ERROR, rank, traceback.format_exc()))
        raise


class Launcher:
    def __init__(self, rc: RunConfig, mc: ModelConfig,
                 entry: Callable, board_path: Optional[str] = None):
        """entry(rank, world, rc, mc, metric_sink, heartbeat) runs one rank."""
        self.rc, self.mc = rc, mc
        self.entry = entry
        self.board = ConsoleBoard(board_path or os.path.join(rc.log_dir, "progress.board"))
        self.epoch_results: Dict[int, List[TrainingIntermediateResult]] = defaultdict(list)
        self.stats: List[EpochStats] = []

    def _spawn(self, ctx, q):
        procs = []
        world = self.rc.num_gpus
        import dataclasses
        rc_dict = dataclasses.asdict(self.rc)
        mc_dict = self.mc.to_dict()
        for rank in range(world):
            p = ctx.Process(target=_rank_main,
                            args=(rank, world, rc_dict, mc_dict, self.entry, q),
                            daemon=False)
            p.start()
            procs.append(p)
        return procs

    def run(self) -> List[EpochStats]:
        attempts = 0
        while True:
            # children inherit this (spawn copies the env): checkpoint saves
            # stamp it so resume can DETECT shard files mixed across a
            # crash+restart generation instead of silently loading them
            os.environ["SHIFU_RUN_ATTEMPT"] = str(attempts)
            ok, err = self._run_once()
            if ok:
                return self.stats
            attempts += 1
            if attempts > self.rc.max_rank_restarts:
                raise RuntimeError(f"training failed after {attempts} attempts: {err}")
            # drop partial metric sets from the crashed attempt: the restarted
            # ranks re-emit the interrupted epoch from scratch, and mixing the
            # two attempts' entries would double-count workers in its stats
            self.epoch_results = defaultdict(
                list, {e: v for e, v in self.epoch_results.items()
                       if len(v) >= self.rc.num_gpus})
            self.board.write(f"[launcher] rank failure ({err}); restarting from "
                             f"latest checkpoint (attempt {attempts})")

    def _run_once(self):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = self._spawn(ctx, q)
        world = self.rc.num_gpus
        last_hb = {r: time.time() for r in range(world)}
        started = {r: False for r in range(world)}   # first message received?
        done = set()
        err: Optional[str] = None
        hb_budget = self.rc.heartbeat_interval_s * self.rc.max_missed_heartbeats
        # until a rank's first message, allow the (long) init grace: model /
        # multi-GB arena construction runs before any heartbeat can be sent
        grace = max(hb_budget, float(getattr(self.rc, "startup_grace_s", 360.0)))

        try:
            while len(done) < world and err is None:
                try:
                    msg, rank, payload = q.get(timeout=self.rc.heartbeat_interval_s)
                    last_hb[rank] = time.time()
                    started[rank] = True
                    if msg == MSG_METRIC:
                        self._on_metric(payload)
                    elif msg == MSG_DONE:
                        done.add(rank)
                    elif msg == MSG_ERROR:
                        err = f"rank {rank} raised:\n{payload}"
                except queue.Empty:
                    pass
                now = time.time()
                for r, p in enumerate(procs):
                    if r in done:
                        continue
                    budget = hb_budget if started[r] else grace
                    if not p.is_alive() and p.exitcode not in (0, None):
                        err = f"rank {r} exited with code {p.exitcode}"
                    elif now - last_hb[r] > budget:
                        err = (f"rank {r} missed heartbeats for {now - last_hb[r]:.0f}s "
                               f"(budget {budget:.0f}s)")
        finally:
            for p in procs:
                if err is not None and p.is_alive():
                    p.terminate()
            for p in procs:
                p.join(timeout=30)
                if p.is_alive():
                    p.kill()
        return err is None, err

    def _on_metric(self, r: TrainingIntermediateResult) -> None:
        self.epoch_results[r.current_epoch].append(r)
        if len(self.epoch_results[r.current_epoch]) == self.rc.num_gpus:
            stats = EpochStats.aggregate(self.epoch_results[r.current_epoch])
            self.stats.append(stats)
            self.board.write(stats.to_console_line())
