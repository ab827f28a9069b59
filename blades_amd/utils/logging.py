"""Two-logger scheme: ``stats`` (one JSON record per line) + ``debug`` (text).

Mirrors the reference's observability contract (reference:
src/blades/utils.py:67-95): a ``stats`` file with one dict per line and a
free-text ``debug`` file under a per-experiment log dir.  Differences,
deliberate:

* records are real JSON (the reference wrote Python-dict ``repr``; its own
  example consumers ``s.replace("'", '"')`` before ``json.loads``, which
  passes real JSON through unchanged, so both readers work);
* the log dir is NOT deleted when it already exists unless
  ``wipe=True`` (the reference always ``shutil.rmtree``'d it --
  reference: src/blades/utils.py:72-74 -- which silently destroyed
  previous results);
* in a multi-rank run only rank 0 attaches file handlers.
"""
from __future__ import annotations

import json
import logging
import os
import shutil
from typing import Any, Dict


def _jsonable(o: Any) -> Any:
    try:
        json.dumps(o)
        return o
    except (TypeError, ValueError):
        return repr(o)


class JsonStatsLogger:
    """Thin wrapper over the ``stats`` logger emitting one JSON object/line."""

    def __init__(self) -> None:
        self._logger = logging.getLogger("stats")

    def write(self, record: Dict[str, Any]) -> None:
        clean = {k: _jsonable(v) for k, v in record.items()}
        self._logger.info(json.dumps(clean))

    # logging.Logger duck-typing: the reference passes dicts to .info()
    def info(self, record: Any) -> None:
        if isinstance(record, dict):
            self.write(record)
        else:
            self._logger.info(record)


def initialize_logger(log_root: str, wipe: bool = False, rank: int = 0) -> None:
    """Create the ``stats`` + ``debug`` file loggers under ``log_root``.

    Reference parity: src/blades/utils.py:67-95.  ``wipe=True`` restores the
    reference's delete-and-recreate behavior.
    """
    if wipe and os.path.exists(log_root):
        shutil.rmtree(log_root)
    os.makedirs(log_root, exist_ok=True)

    for name in ("stats", "debug"):
        logger = logging.getLogger(name)
        logger.setLevel(logging.INFO)
        logger.propagate = False
        # Drop stale file handlers from a previous Simulator in this process.
        for h in list(logger.handlers):
            logger.removeHandler(h)
            h.close()
        if rank == 0:
            fh = logging.FileHandler(os.path.join(log_root, name))
            fh.setLevel(logging.INFO)
            fh.setFormatter(logging.Formatter("%(message)s"))
            logger.addHandler(fh)
        else:  # non-zero ranks log nowhere (avoids file write races)
            logger.addHandler(logging.NullHandler())
