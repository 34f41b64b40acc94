"""Async checkpoint writer (C5 in SURVEY.md §2.9).

The reference wrote every dispatch/upload state to disk synchronously inside
the round (ref:experiment.py:199-202,233-241).  Here `torch.save` runs on a
single background worker so checkpointing overlaps the next round's compute;
ordering guarantees:
  - writes to the same path apply in submission order (single worker);
  - `wait_for(path)` (used by load_state) drains pending writes first, so a
    read always observes the latest submitted state;
  - `flush()` drains everything (called at experiment end / interpreter exit).

Opt-in via FLREID_ASYNC_CKPT=1 (background pickling contends for the GIL,
which hurts CPU-bound runs; on GPU ranks the overlap with device compute
wins).
"""

from __future__ import annotations

import atexit
import os
import queue
import threading
from typing import Any, Optional

import torch


def _to_cpu_snapshot(state: Any) -> Any:
    if torch.is_tensor(state):
        return state.detach().cpu().clone()
    if isinstance(state, dict):
        return {k: _to_cpu_snapshot(v) for k, v in state.items()}
    if isinstance(state, list):
        return [_to_cpu_snapshot(v) for v in state]
    if isinstance(state, tuple):
        return tuple(_to_cpu_snapshot(v) for v in state)
    return state


class AsyncCkptWriter:
    def __init__(self):
        self._q: "queue.Queue" = queue.Queue()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()

    def _ensure_thread(self) -> None:
        with self._lock:
            if self._thread is None or not self._thread.is_alive():
                self._thread = threading.Thread(target=self._worker, daemon=True)
                self._thread.start()

    def _worker(self) -> None:
        while True:
            item = self._q.get()
            if item is None:
                self._q.task_done()
                return
            path, state = item
            try:
                torch.save(state, path)
            except Exception as e:  # pragma: no cover
                print(f"[flreid io] async ckpt write failed for {path}: {e}")
            finally:
                self._q.task_done()

    def submit(self, path: str, state: Any) -> None:
        # snapshot on the caller thread: the live state may mutate next round
        snap = _to_cpu_snapshot(state)
        self._ensure_thread()
        self._q.put((path, snap))

    def flush(self) -> None:
        if self._thread is not None and self._thread.is_alive():
            self._q.join()

    wait_for = flush   # per-path waits degenerate to a drain (single worker)


_WRITER = AsyncCkptWriter()
atexit.register(_WRITER.flush)


def async_enabled() -> bool:
    return os.environ.get("FLREID_ASYNC_CKPT", "0") == "1"


def save_ckpt(path: str, state: Any) -> None:
    if async_enabled():
        _WRITER.submit(path, state)
    else:
        torch.save(state, path)


def before_ckpt_read() -> None:
    """Drain pending writes so reads observe the latest submitted state."""
    if async_enabled():
        _WRITER.flush()
