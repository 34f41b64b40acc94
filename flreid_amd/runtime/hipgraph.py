"""hipGraph capture of launch-bound inner loops.

The FedSTIL head-training step is ~200 small kernel launches per batch
(compose + convs + BN + CE + drift + Adam); on MI355X the round is
launch-bound, not compute-bound.  `GraphedStep` stream-captures one training
step into a hipGraph (torch.cuda.CUDAGraph is hipGraph on ROCm) and replays
it per batch with a device-side copy into static input buffers.

Semantics-preserving protocol: every batch trains exactly once —
  gs = GraphedStep(step_fn)
  gs.warmup(*b0); gs.warmup(*b1)   # real steps, run on the capture-side stream
  gs.capture(*b2)                  # records the graph, then replays it for b2
  gs(*b3); gs(*b4); ...            # replay per batch

Capture policy: recapture per communication round (cheap — the capture costs
about one step) because per-round dispatch re-inits and the optimizer-state
reset would otherwise invalidate captured state; parameter re-inits are
in-place (models/adaptive.py::_assign) so pointers stay stable within a
round.
"""

from __future__ import annotations

import os
from typing import Callable, Sequence, Tuple

import torch


def hipgraph_enabled() -> bool:
    return (torch.cuda.is_available()
            and os.environ.get("FLREID_HIPGRAPH", "1") != "0")


class GraphedStep:
    def __init__(self, fn: Callable):
        self.fn = fn
        self.static_inputs = None
        self.static_outputs = None
        self.graph = None
        self._stream = torch.cuda.Stream()

    def warmup(self, *inputs: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        """Run a REAL step on the side stream (torch's pre-capture warmup
        requirement) — counts as normal training for this batch."""
        if self.static_inputs is None:
            self.static_inputs = [x.clone() for x in inputs]
        else:
            for dst, src in zip(self.static_inputs, inputs):
                dst.copy_(src, non_blocking=True)
        self._stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._stream):
            out = self.fn(*self.static_inputs)
        torch.cuda.current_stream().wait_stream(self._stream)
        return out

    def capture(self, *inputs: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        """Record the graph, then replay it so `inputs` train exactly once."""
        for dst, src in zip(self.static_inputs, inputs):
            dst.copy_(src, non_blocking=True)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_outputs = self.fn(*self.static_inputs)
        self.graph.replay()
        return self.static_outputs

    @property
    def ready(self) -> bool:
        return self.graph is not None

    def __call__(self, *inputs: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        for dst, src in zip(self.static_inputs, inputs):
            dst.copy_(src, non_blocking=True)
        self.graph.replay()
        return self.static_outputs


# ---------------------------------------------------------------------------
# lightweight phase timing (FLREID_PHASE_TIMERS=1): per-phase wall clock with
# a device sync at each boundary — for finding where a round's time goes
# ---------------------------------------------------------------------------

import contextlib
import time
from collections import defaultdict

PHASE_TIMES = defaultdict(float)


def phase_timers_enabled() -> bool:
    return os.environ.get("FLREID_PHASE_TIMERS", "0") == "1"


@contextlib.contextmanager
def phase(name: str):
    if not phase_timers_enabled():
        yield
        return
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        PHASE_TIMES[name] += time.perf_counter() - t0


def dump_phases(reset: bool = True) -> str:
    out = " ".join(f"{k}={v*1000:.0f}ms" for k, v in sorted(PHASE_TIMES.items()))
    if reset:
        PHASE_TIMES.clear()
    return out
