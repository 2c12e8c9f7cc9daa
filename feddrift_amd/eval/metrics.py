"""Metric logging with the reference's metric names.

Keys reproduced from the reference aggregators:
Train/Acc, Train/Loss, Test/Acc, Test/Loss, per-client
{Train,Test}/Acc-CL-i, clustering telemetry Plurality/CL-i, Weight-All/CL-i,
summaries num_models, local_models, Contribute/CL-i, Merge, Reset-m.

Backends: an in-memory record + JSONL file on rank 0; wandb mirrored when
enabled and importable.
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Optional


class MetricLogger:
    def __init__(self, log_dir: str = ".", enabled: bool = True,
                 use_wandb: bool = False, run_name: str = "run",
                 to_file: bool = True):
        self.enabled = enabled
        self.records: List[Dict] = []
        self.summary: Dict = {}
        self._path = os.path.join(log_dir, "metrics.jsonl") \
            if (enabled and to_file) else None
        self._fh = None
        self._wandb = None
        if enabled and use_wandb:
            try:
                import wandb
                self._wandb = wandb
                wandb.init(project="feddrift-amd", name=run_name)
            except Exception:  # noqa: BLE001
                self._wandb = None

    def log(self, metrics: Dict, round_idx: Optional[int] = None) -> None:
        if not self.enabled:
            return
        rec = dict(metrics)
        if round_idx is not None:
            rec["round"] = round_idx
        self.records.append(rec)
        if self._fh is None and self._path is not None:
            self._fh = open(self._path, "a")
        if self._fh is not None:
            self._fh.write(json.dumps(rec) + "\n")
        if self._wandb is not None:
            self._wandb.log(rec)

    def set_summary(self, key: str, value) -> None:
        if not self.enabled:
            return
        self.summary[key] = value
        if self._wandb is not None:
            self._wandb.run.summary[key] = value

    def series(self, key: str) -> List[float]:
        return [r[key] for r in self.records if key in r]

    def mean(self, key: str) -> float:
        s = self.series(key)
        return sum(s) / len(s) if s else float("nan")

    def flush(self) -> None:
        if self._fh is not None:
            self._fh.flush()

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None
        if self._wandb is not None:
            # one wandb run per iteration, like the reference's
            # per-iteration runs (main_fedavg.py:280-290)
            try:
                self._wandb.finish()
            except Exception:  # noqa: BLE001
                pass
            self._wandb = None
