"""hipGraph capture of the steady-state training step.

The reference era relied on framework executors; the MI355X-native
equivalent for a launch-bound inner loop is HIP graph capture (guide:
"capture launch-bound inner loops in hipGraphs").  ``GraphedStep``
captures {zero_grad, forward, loss, backward, finalize, optimizer} once
against static input buffers and replays it per step.

Scope: world_size == 1 by default.  RCCL collectives DO capture and
replay correctly on this stack — proven on hardware at world 1
(tests/test_rccl_gpu.py::test_nccl_ws1_init_and_collectives captures an
all_reduce and replays it) — but the multi-GPU default stays eager
(``DLX_GRAPHS=1`` to force) until an 8-GPU box can validate multi-rank
capture; at the bench's batch sizes the step is not launch-bound, so
eager costs little there.
"""

from __future__ import annotations

from typing import Callable, Optional, Sequence

import torch


class GraphedStep:
    def __init__(self, step_fn: Callable[[Sequence[torch.Tensor]], torch.Tensor],
                 example_inputs: Sequence[torch.Tensor],
                 warmup: int = 3):
        """``step_fn(inputs) -> loss`` must run the FULL training step
        (zero_grad/forward/backward/finalize/opt.step) on the current
        stream using only the given input tensors."""
        self.step_fn = step_fn
        self.static_inputs = [t.clone() for t in example_inputs]
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static_loss: Optional[torch.Tensor] = None
        self._warmup = warmup

    def _capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self._warmup):
                self.step_fn(self.static_inputs)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss = self.step_fn(self.static_inputs)

    def step(self, inputs: Sequence[torch.Tensor]) -> torch.Tensor:
        if self.graph is None:
            self._capture()
        for dst, src in zip(self.static_inputs, inputs):
            dst.copy_(src, non_blocking=True)
        self.graph.replay()
        return self.static_loss


def graphs_enabled(world_size: int) -> bool:
    import os
    v = os.environ.get("DLX_GRAPHS", "auto")
    if v == "0":
        return False
    if v == "1":
        return True
    return world_size == 1  # auto: single-GPU only for now
