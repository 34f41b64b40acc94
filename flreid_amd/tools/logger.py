"""Console logging (ref:tools/logger.py:6-39).

Same surface: per-actor named loggers with `info/warn/error`, plus the two
formatted blocks `info_train` and `info_validation`.  In multi-rank runs each
message is prefixed with the rank so interleaved output stays attributable.
"""

from __future__ import annotations

import logging
import os
import sys

_FORMAT = "%(asctime)s %(levelname).1s %(name)s | %(message)s"
_configured = False


def _ensure_root_config() -> None:
    global _configured
    if _configured:
        return
    handler = logging.StreamHandler(sys.stdout)
    handler.setFormatter(logging.Formatter(_FORMAT, datefmt="%H:%M:%S"))
    root = logging.getLogger("flreid")
    root.setLevel(os.environ.get("FLREID_LOG_LEVEL", "INFO"))
    root.addHandler(handler)
    root.propagate = False
    _configured = True


class Logger:
    """Named logger with the reference's train/validation formatting."""

    def __init__(self, name: str):
        _ensure_root_config()
        rank = os.environ.get("RANK")
        prefix = f"r{rank}." if rank is not None else ""
        self._log = logging.getLogger(f"flreid.{prefix}{name}")
        self.name = name

    def debug(self, msg): self._log.debug(msg)
    def info(self, msg): self._log.info(msg)
    def warn(self, msg): self._log.warning(msg)
    def warning(self, msg): self._log.warning(msg)
    def error(self, msg): self._log.error(msg)

    def info_train(self, task_name, device, data_cnt, acc, loss, epoch=None, total_epoch=None):
        ep = f" epoch {epoch}/{total_epoch}" if epoch is not None else ""
        self.info(
            f"train[{task_name}] on {device}{ep}: "
            f"samples={data_cnt} acc={acc:.4f} loss={loss:.4f}"
        )

    def info_validation(self, task_name, query_size, gallery_size, cmc, mAP):
        r = {k: cmc[k - 1] for k in (1, 3, 5, 10) if len(cmc) >= k}
        ranks = " ".join(f"rank-{k}={v:.4f}" for k, v in r.items())
        self.info(
            f"valid[{task_name}] query={query_size} gallery={gallery_size} "
            f"{ranks} mAP={mAP:.4f}"
        )
