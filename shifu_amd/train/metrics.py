"""Per-epoch training telemetry.

TrainingIntermediateResult carries exactly the reference's metric fields
(reference: core/TrainingIntermediateResult.java:35-45; wire format the
Python side emits at ssgd_monitor.py:288-293:
  worker_index:..,time:..,current_epoch:..,training_loss:..,valid_loss:..,valid_time:..)
but travels over an in-process multiprocessing queue from ranks to the
launcher instead of TCP socket -> ZooKeeper.

EpochStats.aggregate reproduces the AM-side per-epoch statistic
(reference: TensorflowSession.doStatistic:515-549): mean train/valid error
over workers, mean epoch/valid times, and workers sorted by epoch time.
"""
from __future__ import annotations

import json
from dataclasses import asdict, dataclass, field
from typing import List, Optional


@dataclass
class TrainingIntermediateResult:
    worker_index: int = 0
    current_epoch: int = 0
    current_epoch_time: float = 0.0        # seconds spent training this epoch
    current_epoch_valid_time: float = 0.0  # seconds spent on the valid pass
    training_error: float = 0.0
    valid_error: float = 0.0
    container_id: str = ""                 # rank tag (successor of YARN container id)

    def to_line(self) -> str:
        """The reference's key:value,... metric line (SocketServer.java:71-89)."""
        return (f"worker_index:{self.worker_index},time:{self.current_epoch_time},"
                f"current_epoch:{self.current_epoch},training_loss:{self.training_error},"
                f"valid_loss:{self.valid_error},valid_time:{self.current_epoch_valid_time}")

    @classmethod
    def from_line(cls, line: str) -> "TrainingIntermediateResult":
        kv = dict(item.split(":", 1) for item in line.strip().split(","))
        return cls(
            worker_index=int(kv.get("worker_index", 0)),
            current_epoch=int(float(kv.get("current_epoch", 0))),
            current_epoch_time=float(kv.get("time", 0.0)),
            current_epoch_valid_time=float(kv.get("valid_time", 0.0)),
            training_error=float(kv.get("training_loss", 0.0)),
            valid_error=float(kv.get("valid_loss", 0.0)),
        )

    def to_json(self) -> str:
        return json.dumps(asdict(self))


@dataclass
class EpochStats:
    epoch: int
    mean_training_error: float
    mean_valid_error: float
    mean_epoch_time: float
    mean_valid_time: float
    workers_by_time: List[int] = field(default_factory=list)  # slowest last

    @classmethod
    def aggregate(cls, results: List[TrainingIntermediateResult]) -> "EpochStats":
        if not results:
            raise ValueError("no results to aggregate")
        n = len(results)
        by_time = sorted(results, key=lambda r: r.current_epoch_time)
        return cls(
            epoch=results[0].current_epoch,
            mean_training_error=sum(r.training_error for r in results) / n,
            mean_valid_error=sum(r.valid_error for r in results) / n,
            mean_epoch_time=sum(r.current_epoch_time for r in results) / n,
            mean_valid_time=sum(r.current_epoch_valid_time for r in results) / n,
            workers_by_time=[r.worker_index for r in by_time],
        )

    def to_console_line(self) -> str:
        """One line for the progress board (successor of the HDFS
        ClientConsoleBoard, CommonUtils.java:426-458)."""
        return (f"epoch {self.epoch}: train_err={self.mean_training_error:.6f} "
                f"valid_err={self.mean_valid_error:.6f} "
                f"epoch_time={self.mean_epoch_time:.3f}s valid_time={self.mean_valid_time:.3f}s "
                f"workers_by_time={self.workers_by_time}")


class ConsoleBoard:
    """Append-only progress file the launcher tails to the console
    (successor of CommonUtils.ClientConsoleBoard on HDFS)."""

    def __init__(self, path: Optional[str] = None, echo: bool = True):
        self.path = path
        self.echo = echo
        if path:
            import os
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            open(path, "a").close()

    def write(self, line: str) -> None:
        if self.path:
            with open(self.path, "a") as f:
                f.write(line + "\n")
        if self.echo:
            print(line, flush=True)
