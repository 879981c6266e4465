"""hipGraph-captured training step.

MI355X-first: a Wide&Deep step is ~50 short kernels; capturing the whole
fwd+loss+bwd+all-reduce+optimizer sequence as ONE hipGraph removes the
per-kernel launch overhead (torch.cuda.CUDAGraph is hipGraph on ROCm).

Requirements honored by the framework:
* static shapes (tabular batches are fixed-size; sparse embedding grads have
  nnz == B*F because the sparse path never dedups — ops/embedding.py);
* no host syncs inside the captured closure (loss is returned as a device
  tensor; Adam bias correction uses a device-side step counter —
  ops/optim.py adam_step_dev).

Usage:
    stepper = GraphedStep(step_fn, warmup=3)   # step_fn() -> loss tensor
    loss = stepper.run()                       # replay
Inputs must live in static buffers the caller copies into before run().
"""
from __future__ import annotations

from typing import Callable, Optional

import torch


class GraphedStep:
    def __init__(self, step_fn: Callable[[], torch.Tensor], warmup: int = 3):
        self.step_fn = step_fn
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static_loss: Optional[torch.Tensor] = None
        self._warmup = warmup

    def capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self._warmup):
                self.step_fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss = self.step_fn()

    def run(self) -> torch.Tensor:
        if self.graph is None:
            self.capture()
        self.graph.replay()
        return self.static_loss
