"""Structured JSON experiment log (ref:experiment.py:16-55).

Same dotted-key nested-dict semantics (`data.client-0.3.task-0-1`) with
append-to-list / set-add / dict-merge on key collision
(ref:experiment.py:23-42).  Differences from the reference:
  - the JSON file is written per flush (once per round), not on every record
    (the reference rewrote the whole file every record);
  - multi-rank: every rank records locally; `sync(ctx)` gathers pending
    records and rank 0 merges + writes one JSON, so the output file matches
    the single-process layout byte-for-byte in content.
"""

from __future__ import annotations

import json
import os
import threading
from typing import Any, Dict, List, Tuple


def _jsonable(value: Any):
    import torch

    if isinstance(value, torch.Tensor):
        return value.tolist()
    if isinstance(value, dict):
        return {k: _jsonable(v) for k, v in value.items()}
    if isinstance(value, (list, tuple)):
        return [_jsonable(v) for v in value]
    if isinstance(value, set):
        return sorted(_jsonable(v) for v in value)
    if hasattr(value, "item") and callable(value.item) and getattr(value, "ndim", None) == 0:
        return value.item()
    return value


class ExperimentLog:
    def __init__(self, save_path: str):
        self.records: Dict = {}
        self.save_path = save_path
        self.pending: List[Tuple[str, Any]] = []
        self._lock = threading.Lock()

    def _update_iter(self, key: str, value: Any) -> None:
        keys = key.split(".")
        node = self.records
        for k in keys[:-1]:
            node = node.setdefault(k, {})
        leaf = keys[-1]
        if leaf not in node:
            node[leaf] = value
        else:
            cur = node[leaf]
            if isinstance(cur, list):
                cur.append(value)
            elif isinstance(cur, set):
                cur.add(value)
            elif isinstance(cur, dict):
                cur.update(value)
            else:
                node[leaf] = value

    def record(self, key: str, value: Any) -> None:
        value = _jsonable(value)
        with self._lock:
            self._update_iter(key, value)
            self.pending.append((key, value))

    def apply_remote(self, pending: List[Tuple[str, Any]]) -> None:
        with self._lock:
            for key, value in pending:
                self._update_iter(key, value)

    def flush(self) -> None:
        dirname = os.path.dirname(self.save_path)
        if dirname:
            os.makedirs(dirname, exist_ok=True)
        with self._lock:
            with open(self.save_path, "w") as f:
                json.dump(self.records, f, indent=2, default=str)

    def sync(self, ctx) -> None:
        """Gather pending records across ranks; rank 0 merges + writes."""
        if ctx is None or not ctx.is_distributed:
            self.flush()
            self.pending.clear()
            return
        gathered = ctx.all_gather_object(self.pending)
        if ctx.is_rank0():
            for rank, pend in enumerate(gathered):
                if rank != ctx.rank:
                    self.apply_remote(pend)
            self.flush()
        self.pending.clear()

    def maybe_sync(self, ctx, curr_round: int, total_rounds: int) -> None:
        """Round-batched sync: the cross-rank object gather runs every
        FLREID_LOG_SYNC_EVERY rounds (default 10) and on the final round,
        not per round — the per-round O(world_size) serialisation would
        otherwise sit on the 8-rank critical path.  Pending records
        accumulate locally in between (the decision is a pure function of
        (round, interval), so every rank takes the collective together)."""
        if ctx is None or not ctx.is_distributed:
            self.flush()
            self.pending.clear()
            return
        interval = max(1, int(os.environ.get("FLREID_LOG_SYNC_EVERY", "10")))
        if curr_round % interval == 0 or curr_round >= total_rounds:
            self.sync(ctx)
