"""Run-directory negotiation and scalar logging.

Parity with sheeprl/utils/logger.py:12-89: rank 0 creates
``logs/runs/<root_dir>/<run_name>/version_N`` and broadcasts it to all ranks.
TensorBoard/MLflow are not in this image, so the default logger writes scalars
to ``metrics.jsonl`` (one JSON object per log call) plus a final ``metrics.csv``
— enough for dashboards and the tests; the writer API mirrors what the algos
need (``log_metrics(dict, step)``).
"""

from __future__ import annotations

import csv
import json
import os
from pathlib import Path
from typing import Any, Dict, List, Optional


class JsonlLogger:
    def __init__(self, log_dir: str | Path, tensorboard: bool = True) -> None:
        self.log_dir = Path(log_dir)
        self.log_dir.mkdir(parents=True, exist_ok=True)
        self._path = self.log_dir / "metrics.jsonl"
        self._fh = open(self._path, "a", buffering=1)
        self._keys: List[str] = []
        # parity with the reference's TensorBoard default logger
        # (sheeprl/utils/logger.py:12): the in-repo tfevents writer needs no
        # tensorboard package (utils/tboard.py)
        self._tb = None
        if tensorboard:
            from sheeprl_amd.utils.tboard import TensorBoardWriter

            self._tb = TensorBoardWriter(str(self.log_dir))

    @property
    def name(self) -> str:
        return "jsonl"

    def log_metrics(self, metrics: Dict[str, Any], step: Optional[int] = None) -> None:
        rec = {"step": step}
        for k, v in metrics.items():
            try:
                rec[k] = float(v)
            except (TypeError, ValueError):
                rec[k] = str(v)
        self._fh.write(json.dumps(rec) + "\n")
        if self._tb is not None:
            self._tb.add_scalars(
                {k: v for k, v in rec.items() if k != "step" and isinstance(v, float)},
                int(step or 0),
            )

    def log_hyperparams(self, params: Dict[str, Any]) -> None:
        with open(self.log_dir / "hparams.json", "w") as f:
            json.dump(params, f, indent=2, default=str)

    def finalize(self) -> None:
        self._fh.flush()
        try:
            rows = [json.loads(line) for line in open(self._path)]
        except Exception:
            return
        keys: List[str] = []
        for r in rows:
            for k in r:
                if k not in keys:
                    keys.append(k)
        with open(self.log_dir / "metrics.csv", "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=keys)
            w.writeheader()
            w.writerows(rows)

    def close(self) -> None:
        self.finalize()
        self._fh.close()
        if self._tb is not None:
            self._tb.close()


def get_log_dir(runtime: Any, root_dir: str, run_name: str, share: bool = True) -> str:
    """Create (rank 0) and share the versioned run directory across ranks."""
    base = Path("logs") / "runs" / root_dir / run_name
    if runtime.global_rank == 0:
        version = 0
        while (base / f"version_{version}").exists():
            version += 1
        log_dir = base / f"version_{version}"
        log_dir.mkdir(parents=True, exist_ok=True)
        payload = [str(log_dir)]
    else:
        payload = [None]
    if share and runtime.world_size > 1:
        payload = runtime.broadcast_object_list(payload, src=0)
    return payload[0]


def get_logger(runtime: Any, cfg: Any, log_dir: str | Path) -> Optional[JsonlLogger]:
    if runtime.global_rank != 0:
        return None
    if cfg.metric.log_level <= 0:
        return None
    tb = bool(cfg.metric.get("tensorboard", True)) if hasattr(cfg.metric, "get") else True
    return JsonlLogger(log_dir, tensorboard=tb)
