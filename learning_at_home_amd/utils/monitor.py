"""Lightweight observability (SURVEY.md §5 "Metrics / logging"): per-stage
wall/GPU timers, expert-load statistics, and an amd-smi/xGMI snapshot helper.
No external services — structured dict output, printable or JSON-dumpable."""
from __future__ import annotations

import json
import subprocess
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, Optional

import torch


class StageTimers:
    """Accumulating per-stage timers. On GPU, brackets stages with CUDA/HIP
    events so timings reflect device time, not launch time."""

    def __init__(self, use_gpu_events: Optional[bool] = None):
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self.use_gpu = torch.cuda.is_available() if use_gpu_events is None else use_gpu_events
        self._pending = []

    @contextmanager
    def __call__(self, stage: str):
        if self.use_gpu:
            start, end = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
            start.record()
            yield
            end.record()
            self._pending.append((stage, start, end))
        else:
            t0 = time.perf_counter()
            yield
            self.totals[stage] += time.perf_counter() - t0
            self.counts[stage] += 1

    def flush(self) -> None:
        """Resolve pending GPU events (synchronizes)."""
        if self._pending:
            torch.cuda.synchronize()
            for stage, start, end in self._pending:
                self.totals[stage] += start.elapsed_time(end) / 1e3
                self.counts[stage] += 1
            self._pending.clear()

    def summary(self) -> Dict[str, dict]:
        self.flush()
        return {
            stage: {"total_s": round(t, 4), "count": self.counts[stage],
                    "mean_ms": round(1e3 * t / max(1, self.counts[stage]), 3)}
            for stage, t in sorted(self.totals.items(), key=lambda kv: -kv[1])
        }

    def report(self) -> str:
        return json.dumps(self.summary())


def expert_load_stats(counts: torch.Tensor, capacity: int) -> dict:
    """Routing balance metrics from a per-expert assignment count vector."""
    c = counts.float()
    total = float(c.sum())
    mean = total / c.numel()
    overflow = float((c - capacity).clamp(min=0).sum())
    p = c / max(total, 1.0)
    entropy = float(-(p.clamp(min=1e-12) * p.clamp(min=1e-12).log()).sum())
    return {
        "experts": c.numel(),
        "mean_load": round(mean, 1),
        "max_load": int(c.max()),
        "min_load": int(c.min()),
        "cv": round(float(c.std(unbiased=False)) / max(mean, 1e-9), 3),
        "dropped_frac": round(overflow / max(total, 1.0), 4),
        "entropy_frac": round(entropy / float(torch.log(torch.tensor(float(c.numel())))), 4),
    }


def amd_smi_snapshot() -> dict:
    """Best-effort GPU utilization/VRAM/xGMI snapshot via rocm-smi."""
    out = {}
    try:
        raw = subprocess.run(
            ["/opt/rocm/bin/rocm-smi", "--showuse", "--showmeminfo", "vram", "--json"],
            capture_output=True, text=True, timeout=10,
        ).stdout
        out["rocm_smi"] = json.loads(raw) if raw.strip().startswith("{") else raw[-2000:]
    except Exception as e:  # noqa: BLE001
        out["rocm_smi_error"] = repr(e)
    try:  # xGMI link state/throughput (multi-GPU boxes; absent counters ignored)
        raw = subprocess.run(
            ["/opt/rocm/bin/rocm-smi", "--showtopo", "--json"],
            capture_output=True, text=True, timeout=10,
        ).stdout
        out["xgmi_topo"] = json.loads(raw) if raw.strip().startswith("{") else raw[-2000:]
    except Exception as e:  # noqa: BLE001
        out["xgmi_error"] = repr(e)
    return out
