"""Observability: per-role metric runs + system metrics.

Reimplements the reference's MLflow subsystem
(/root/reference/hivetrain/utils/mlflow_utils.py) without requiring an
MLflow server: metrics stream to a JSONL file per run (rocprof-friendly,
greppable, no network), and mirror to MLflow automatically when the
``mlflow`` package is importable AND a tracking URI is configured — the
reference's own default is disabled too (MLFLOW_ACTIVE=False,
config/mlflow_config.py:3).

Parity map:
  initialize_mlflow(role, ...)   -> MetricsRun(role, hotkey, ...)
  log_model_metrics(step, **kw)  -> run.log(step, **kw)
  get_gpu/cpu/memory/network_usage (mlflow_utils.py:15-69)
                                 -> system_metrics()
  VERSION regex over template/__init__.py (mlflow_utils.py:72-82)
                                 -> framework __version__ / __spec_version__
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional

import torch


def _try_mlflow():
    uri = os.environ.get("MLFLOW_TRACKING_URI")
    if not uri:
        return None
    try:
        import mlflow  # type: ignore
        mlflow.set_tracking_uri(uri)
        return mlflow
    except ImportError:
        return None


def gpu_utilization() -> Optional[float]:
    """Device utilization percent (reference: torch.cuda.utilization,
    mlflow_utils.py:15-30); None without a GPU."""
    if not torch.cuda.is_available():
        return None
    try:
        return float(torch.cuda.utilization())
    except Exception:
        return None


def gpu_memory_gb() -> Optional[float]:
    if not torch.cuda.is_available():
        return None
    return torch.cuda.memory_allocated() / 1e9


def system_metrics() -> Dict[str, float]:
    """CPU %, RSS, net bytes (reference: mlflow_utils.py:33-69), plus GPU
    util/mem when present."""
    out: Dict[str, float] = {}
    try:
        import psutil
        p = psutil.Process()
        out["cpu_percent"] = psutil.cpu_percent(interval=None)
        out["rss_gb"] = p.memory_info().rss / 1e9
        net = psutil.net_io_counters()
        out["net_sent_mb"] = net.bytes_sent / 1e6
        out["net_recv_mb"] = net.bytes_recv / 1e6
    except ImportError:  # pragma: no cover
        pass
    g = gpu_utilization()
    if g is not None:
        out["gpu_util"] = g
        out["gpu_mem_gb"] = gpu_memory_gb() or 0.0
    return out


class MetricsRun:
    """One metrics stream for one role process.

    Run naming matches the reference (miner_{hotkey} / validator_{hotkey} /
    AVERAGER, mlflow_utils.py:85-123). Writes one JSON object per line:
    {"ts": ..., "step": ..., "run": ..., <metrics>}.
    """

    def __init__(self, role: str, hotkey: str = "", log_dir: str = "metrics",
                 hyperparams: Optional[dict] = None,
                 system_every: int = 50):
        from .. import __version__, __spec_version__
        self.run_name = ("AVERAGER" if role == "averager"
                         else f"{role}_{hotkey}" if hotkey else role)
        self.system_every = system_every
        self._n = 0
        os.makedirs(log_dir, exist_ok=True)
        self.path = os.path.join(log_dir, f"{self.run_name}.jsonl")
        self._f = open(self.path, "a", buffering=1)
        self._mlflow = _try_mlflow()
        if self._mlflow is not None:  # pragma: no cover - needs server
            self._mlflow.set_experiment(role)
            self._mlflow.start_run(run_name=self.run_name)
        self.log_params({"version": __version__,
                         "spec_version": __spec_version__,
                         **(hyperparams or {})})

    def log_params(self, params: dict) -> None:
        self._write({"event": "params", **params})
        if self._mlflow is not None:  # pragma: no cover
            self._mlflow.log_params(params)

    def log(self, step: int, **metrics: float) -> None:
        """reference: log_model_metrics (mlflow_utils.py:126-140)."""
        if self._n % self.system_every == 0:
            metrics = {**metrics, **system_metrics()}
        self._n += 1
        self._write({"step": step, **metrics})
        if self._mlflow is not None:  # pragma: no cover
            self._mlflow.log_metrics(
                {k: v for k, v in metrics.items()
                 if isinstance(v, (int, float))}, step=step)

    def _write(self, obj: dict) -> None:
        obj = {"ts": time.time(), "run": self.run_name, **obj}
        self._f.write(json.dumps(obj) + "\n")

    def close(self) -> None:
        self._f.close()
        if self._mlflow is not None:  # pragma: no cover
            self._mlflow.end_run()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
