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


_EPOCH_GRAPH_AUTO = False


def suggest_epoch_graph(comm_rounds: int) -> None:
    """Auto-enable whole-epoch capture for LONG runs: the ~1 s one-time
    instantiation amortises against the measured −4 ms/round at ≈250 rounds.
    An explicit FLREID_EPOCH_GRAPH always wins."""
    global _EPOCH_GRAPH_AUTO
    _EPOCH_GRAPH_AUTO = comm_rounds >= 300


def epoch_graph_enabled() -> bool:
    """Whole-epoch capture: opt-in via FLREID_EPOCH_GRAPH=1, opted out via
    =0, otherwise auto-on for runs long enough to amortise the capture
    (suggest_epoch_graph).  Measured: steady-state −4 ms/round over the
    per-step graphs, ~1 s one-time ~8000-node graph instantiation."""
    if not hipgraph_enabled():
        return False
    explicit = os.environ.get("FLREID_EPOCH_GRAPH")
    if explicit is not None:
        return explicit == "1"
    return _EPOCH_GRAPH_AUTO


class EpochGraph:
    """Captures an ENTIRE rehearsal epoch — S full training steps over a
    device-resident sample store — as ONE hipGraph.

    Per-epoch inputs are data, not topology: the store content, the target
    ids, and the [S·B] shuffled batch-index matrix are copied into static
    buffers, then one replay executes all S steps (index_select → forward →
    backward → Adam, ~200 kernels each) back-to-back with zero host
    involvement between steps.  The per-step GraphedStep already removed the
    CPU launch cost; this removes the remaining per-step replay/submission
    gaps and the out-of-graph batch gathers.

    Same train-exactly-once protocol as GraphedStep: two warmup epochs run
    eagerly, the third records the graph and immediately replays it, later
    epochs replay.  Metric outputs are two device scalars (Σloss, Σacc)
    accumulated inside the graph — one host sync per epoch at most.
    """

    def __init__(self, step_fn: Callable, steps: int, batch: int,
                 store_template: torch.Tensor, pids_template: torch.Tensor,
                 present: Callable):
        self.step_fn = step_fn
        self.S, self.B = steps, batch
        self.store = torch.empty_like(store_template)
        self.pids = torch.empty_like(pids_template)
        self.idx = torch.empty(steps * batch, dtype=torch.long,
                               device=store_template.device)
        self.present = present
        self.graph = None
        self.static_out = None
        self._warmups = 0
        self._stream = torch.cuda.Stream()

    def _epoch_body(self):
        loss_t = acc_t = None
        idx2 = self.idx.view(self.S, self.B)
        for s in range(self.S):
            data = self.present(self.store.index_select(0, idx2[s]))
            target = self.pids.index_select(0, idx2[s])
            loss, acc = self.step_fn(data, target)
            loss_t = loss.clone() if loss_t is None else loss_t + loss
            acc_t = acc.clone() if acc_t is None else acc_t + acc
        return loss_t, acc_t

    def run(self, store: torch.Tensor, pids: torch.Tensor,
            idx_flat: torch.Tensor):
        self.store.copy_(store, non_blocking=True)
        self.pids.copy_(pids, non_blocking=True)
        self.idx.copy_(idx_flat, non_blocking=True)
        if self.graph is not None:
            self.graph.replay()
            return self.static_out
        if self._warmups < 1:
            self._warmups += 1
            self._stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._stream):
                out = self._epoch_body()
            torch.cuda.current_stream().wait_stream(self._stream)
            return out
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_out = self._epoch_body()
        self.graph.replay()
        return self.static_out


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
