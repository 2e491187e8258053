"""Per-rank timer logging with the reference's key names.

Mirrors arrow/common/wb_logging.py: `log`/`set_iteration_data`/`finish`/
`wandb_init` with the same timing keys (`spmm_kernel_time`, `spmm_time`,
`spmm_x_bcast_time`, `spmm_row_reduce`, `back_agg_*`, `forward_agg_*`,
`init_time`, ...) so downstream tooling matches. wandb itself is optional
(absent in this environment); logs fall back to pickle/txt under ./logs
(reference wb_logging.py:67-114).
"""
import os
import pickle
import uuid
from pathlib import Path
from typing import Optional

_LOGS = []
_ITERATION_DATA = {}
_CONFIG = {}
_COMM = None
_ENABLED = False


def set_iteration_data(data: dict):
    global _ITERATION_DATA
    _ITERATION_DATA = dict(data)


def log(data: dict):
    if not _ENABLED:
        return
    d = dict(data)
    d.update(_ITERATION_DATA)
    _LOGS.append(d)


def drain():
    """Return and clear the accumulated log entries (used by bench timers)."""
    global _LOGS
    out, _LOGS = _LOGS, []
    return out


def wandb_init(comm, dataset, n_features, iterations, device, algorithm,
               block_width, wandb_api_key: Optional[str] = None):
    """Keeps the reference signature (wb_logging.py:163-205)."""
    global _CONFIG, _COMM, _ENABLED
    _COMM = comm
    _ENABLED = True
    dataset_name = (dataset.split('/'))[-1] if dataset is not None else "synthetic"
    _CONFIG = {
        "dataset": dataset_name,
        "width": block_width,
        "n_features": n_features,
        "iterations": iterations,
        "device": device,
        "ranks": getattr(comm, 'size', 1),
        "host": "NA",
        "algorithm": algorithm,
    }
    set_iteration_data({})
    return None


def finish():
    """Write this rank's logs to ./logs (rank 0 only), reference
    wb_logging.py:83-114 file fallback."""
    global _ENABLED
    if not _ENABLED:
        return
    _ENABLED = False
    if _COMM is not None and getattr(_COMM, 'rank', 0) != 0:
        return
    if not _LOGS:
        return
    algorithm = _CONFIG.get("algorithm", "arrow_amd")
    dataset = _CONFIG.get("dataset", "unknown")
    run_id = f"{algorithm}.{dataset}." + str(uuid.uuid1())
    base = Path("./logs")
    base.mkdir(parents=True, exist_ok=True)
    with open(base / f"{run_id}.pickle", "wb") as f:
        pickle.dump(_LOGS, f)
    with open(base / f"{run_id}.txt", "w") as f:
        f.write(str(_LOGS))
    with open(base / f"{run_id}.config", "w") as f:
        f.write(str(_CONFIG))
    with open(base / f"{run_id}.config.pickle", "wb") as f:
        pickle.dump(_CONFIG, f)
