"""Observability: scalar logging + HPO reporting.

Parity targets: the reference logs scalars to TensorBoard
(MyTensorBoardLogger / SummaryWriter) and reports intermediate/final F1 to
NNI for HPO (base_module.py:346, main_cli.py:184). Neither tensorboard nor
nni are importable in this environment, so:

  * ScalarLogger writes scalars to a JSONL file (`scalars.jsonl`) and
    mirrors them to TensorBoard when the package is available;
  * HPOReporter appends intermediate metrics and the final metric to
    `hpo_metrics.jsonl`, and calls nni.report_* when nni is importable —
    the same injection surface as main_cli's --hpo_params.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional


class ScalarLogger:
    def __init__(self, log_dir: str, name: str = "scalars.jsonl"):
        os.makedirs(log_dir, exist_ok=True)
        self.path = os.path.join(log_dir, name)
        self._tb = None
        try:  # optional TensorBoard mirror
            from torch.utils.tensorboard import SummaryWriter

            self._tb = SummaryWriter(log_dir)
        except Exception:
            self._tb = None

    def log(self, scalars: Dict[str, float], step: int):
        row = {"step": step, "ts": time.time()}
        row.update({k: float(v) for k, v in scalars.items() if isinstance(v, (int, float))})
        with open(self.path, "a") as f:
            f.write(json.dumps(row) + "\n")
        if self._tb is not None:
            for k, v in scalars.items():
                if isinstance(v, (int, float)):
                    self._tb.add_scalar(k, v, step)

    def close(self):
        if self._tb is not None:
            self._tb.close()


class HPOReporter:
    def __init__(self, out_dir: str):
        os.makedirs(out_dir, exist_ok=True)
        self.path = os.path.join(out_dir, "hpo_metrics.jsonl")
        try:
            import nni  # noqa: F401

            self._nni = nni
        except ImportError:
            self._nni = None

    def report_intermediate(self, value: float):
        with open(self.path, "a") as f:
            f.write(json.dumps({"intermediate": float(value)}) + "\n")
        if self._nni:
            self._nni.report_intermediate_result(value)

    def report_final(self, value: float):
        with open(self.path, "a") as f:
            f.write(json.dumps({"final": float(value)}) + "\n")
        if self._nni:
            self._nni.report_final_result(value)

    @staticmethod
    def get_next_parameter() -> Optional[Dict]:
        """NNI parameter injection equivalent: reads the file named by
        DEEPDFA_HPO_PARAMS (a JSON dict of dotted config overrides)."""
        path = os.environ.get("DEEPDFA_HPO_PARAMS")
        if path and os.path.exists(path):
            with open(path) as f:
                return json.load(f)
        return None
