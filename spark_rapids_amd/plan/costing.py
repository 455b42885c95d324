"""Cost-based optimizer (reference analogue: CostBasedOptimizer.scala —
CpuCostModel/GpuCostModel with per-operator costs and transfer rates,
default OFF, veto power over GPU conversion).

Estimates row counts bottom-up (parquet footers give exact scan
cardinality; filters/joins/aggregates use the reference's style of
coarse selectivity factors), prices the plan on both backends (per-row
operator costs + host->device transfer of the scanned bytes), and vetoes
GPU placement when the estimated GPU cost (including transfer) exceeds
the CPU cost — the "don't accelerate tiny sections" decision.
"""
from __future__ import annotations

from typing import Tuple

from . import logical as L

# per-row operator costs, mirroring the reference's defaults
# (RapidsConf: spark.rapids.sql.optimizer.defaultCpuOperatorCost 0.0002,
# defaultGpuOperatorCost 0.0001) and its transfer-speed model
CPU_OP_COST = 0.0002 / 1000       # seconds per row per operator
GPU_OP_COST = 0.0001 / 1000
TRANSFER_BYTES_PER_SEC = 32e9      # pinned H2D
GPU_FIXED_OVERHEAD_S = 0.001       # launches/sync per operator


def estimate_rows(node: L.LogicalPlan) -> float:
    if isinstance(node, L.Scan):
        src = node.source
        files = getattr(src, "files", None)
        if files:
            try:
                from ..io.parquet_gpu import _file_meta

                return float(sum(_file_meta(f)[0].num_rows for f in files))
            except Exception:  # noqa: BLE001 - non-parquet source
                pass
        batches = getattr(src, "batches", None)
        if batches is not None:
            return float(sum(b.num_rows for b in batches))
        return 1e6
    kids = [estimate_rows(c) for c in node.children]
    if isinstance(node, L.Filter):
        return kids[0] * 0.25
    if isinstance(node, (L.Join, L.CrossJoin)):
        return max(kids) if kids else 0.0
    if isinstance(node, L.Aggregate):
        return min(kids[0], max(kids[0] ** 0.5, 1.0)) if kids else 1.0
    if isinstance(node, L.Limit):
        return min(kids[0], float(node.n))
    if isinstance(node, L.Union):
        return sum(kids)
    return kids[0] if kids else 1.0


def _scan_bytes(node: L.LogicalPlan) -> float:
    total = 0.0
    if isinstance(node, L.Scan):
        files = getattr(node.source, "files", None)
        if files:
            import os

            try:
                total += sum(os.path.getsize(f) for f in files)
            except OSError:
                pass
        batches = getattr(node.source, "batches", None)
        if batches is not None:
            total += sum(b.nbytes for b in batches)
    for c in node.children:
        total += _scan_bytes(c)
    return total


def _plan_cost(node: L.LogicalPlan, per_row: float) -> Tuple[float, int]:
    rows = estimate_rows(node)
    cost = rows * per_row
    n_ops = 1
    for c in node.children:
        cc, cn = _plan_cost(c, per_row)
        cost += cc
        n_ops += cn
    return cost, n_ops


def evaluate(plan: L.LogicalPlan):
    """-> (keep_on_gpu: bool, note: str) — the conversion veto."""
    cpu_cost, n_ops = _plan_cost(plan, CPU_OP_COST)
    gpu_compute, _ = _plan_cost(plan, GPU_OP_COST)
    transfer = _scan_bytes(plan) / TRANSFER_BYTES_PER_SEC
    gpu_cost = gpu_compute + transfer + n_ops * GPU_FIXED_OVERHEAD_S
    keep = gpu_cost < cpu_cost
    note = (f"cost-based optimizer: est cpu={cpu_cost * 1e3:.3f}ms "
            f"gpu={gpu_cost * 1e3:.3f}ms (transfer "
            f"{transfer * 1e3:.3f}ms, {n_ops} ops)")
    return keep, note
