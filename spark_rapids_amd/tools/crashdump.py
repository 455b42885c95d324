"""Fatal-GPU-error diagnostics (reference analogue: GpuCoreDumpHandler +
the executor fail-fast path, Plugin.scala:727-737 — on CudaException the
executor logs GPU debug info before exiting so the scheduler reschedules).

Here: when a query dies with a HIP-level error, write a diagnostic bundle
(device + pool state, task metrics, the failing plan) under
$RAPIDS_CRASH_DIR (default /tmp/rapids_crash) so a dead run is
debuggable post-mortem, then re-raise.
"""
from __future__ import annotations

import json
import os
import subprocess
import time
import traceback
from typing import Optional

_HIP_MARKERS = ("HIP", "hip error", "device-side", "out of memory",
                "Memory access fault", "CUDA")


def _is_gpu_error(exc: BaseException) -> bool:
    msg = f"{type(exc).__name__}: {exc}"
    return isinstance(exc, (RuntimeError, MemoryError)) and \
        any(m in msg for m in _HIP_MARKERS)


def dump(exc: BaseException, plan_text: Optional[str] = None) -> Optional[str]:
    """Write the crash bundle; returns its path (None if disabled or the
    error is not GPU-related)."""
    if os.environ.get("RAPIDS_CRASH_DUMP", "1") == "0":
        return None
    if not _is_gpu_error(exc):
        return None
    d = os.environ.get("RAPIDS_CRASH_DIR", "/tmp/rapids_crash")
    try:
        os.makedirs(d, exist_ok=True)
        path = os.path.join(d, f"crash_{int(time.time())}_{os.getpid()}")
        info = {"error": f"{type(exc).__name__}: {exc}",
                "traceback": traceback.format_exc()}
        try:
            from ..metrics import task_metrics

            info["task_metrics"] = task_metrics()
        except Exception:  # noqa: BLE001
            pass
        try:
            from ..memory import device_pool

            info["device_pool"] = device_pool.stats()
        except Exception:  # noqa: BLE001
            pass
        try:
            import torch

            if torch.cuda.is_available():
                info["device"] = torch.cuda.get_device_name(0)
        except Exception:  # noqa: BLE001
            pass
        if plan_text:
            info["plan"] = plan_text
        try:
            smi = subprocess.run(
                ["rocm-smi", "--showmeminfo", "vram", "--showuse"],
                capture_output=True, text=True, timeout=10)
            info["rocm_smi"] = smi.stdout[-4000:]
        except Exception:  # noqa: BLE001
            pass
        with open(path + ".json", "w") as f:
            json.dump(info, f, indent=2, default=str)
        return path + ".json"
    except OSError:
        return None
