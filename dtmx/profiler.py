"""Profiler (reference src/profiler/, python/mxnet/profiler.py).

Chrome-trace output via torch.profiler (rocTracer underneath on ROCm) —
`set_config(filename=...)` + `set_state('run'|'stop')` mirrors
mx.profiler. The reference's "profile the server remotely" command
(kvstore_dist.h:102-110) maps to profiling any rank: every dtmx rank is both
worker and (shard-)server, so `rank_filter` selects which ranks trace.
"""
from __future__ import annotations

import logging
import os
from typing import Optional

import torch

_config = {"filename": "profile.json", "profile_all": True, "rank_filter": None}
_prof: Optional[torch.profiler.profile] = None


def set_config(filename: str = "profile.json", profile_all: bool = True,
               rank_filter=None, **kwargs):
    _config.update(filename=filename, profile_all=profile_all, rank_filter=rank_filter)


def set_state(state: str = "stop"):
    global _prof
    if state == "run":
        rank = int(os.environ.get("RANK", "0"))
        rf = _config["rank_filter"]
        if rf is not None and rank not in rf:
            return
        activities = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(torch.profiler.ProfilerActivity.CUDA)
        _prof = torch.profiler.profile(activities=activities)
        _prof.__enter__()
    elif state == "stop":
        if _prof is not None:
            _prof.__exit__(None, None, None)
            _prof.export_chrome_trace(_config["filename"])
            logging.info("profiler trace written to %s", _config["filename"])


def dumps(reset: bool = False, sort_by: str = "self_device_time",
          max_rows: int = 40) -> str:
    """Aggregate per-op statistics table (reference profiler
    aggregate_stats.cc / mx.profiler.dumps): op name, call count, total and
    mean device/host time. Call between set_state('run') and 'stop', or
    after 'stop' on the captured events."""
    if _prof is None:
        return "(profiler not running; set_state('run') first)"
    events = _prof.key_averages()
    key = {"self_device_time": "self_device_time_total",
           "device_time": "device_time_total",
           "cpu_time": "self_cpu_time_total"}.get(sort_by, sort_by)
    rows = sorted(events, key=lambda e: -getattr(e, key, 0))[:max_rows]
    out = [f"{'op':60s} {'calls':>7s} {'dev_ms':>10s} {'mean_us':>9s} "
           f"{'cpu_ms':>10s}"]
    for e in rows:
        dev_ms = getattr(e, "self_device_time_total", 0) / 1e3
        cpu_ms = e.self_cpu_time_total / 1e3
        mean_us = (getattr(e, "self_device_time_total", 0) / e.count
                   if e.count else 0.0)
        out.append(f"{e.key[:60]:60s} {e.count:7d} {dev_ms:10.3f} "
                   f"{mean_us:9.1f} {cpu_ms:10.3f}")
    return "\n".join(out)


def dump(aggregate: bool = False):
    """Stop and write the chrome trace; with aggregate=True also log the
    per-op summary table (reference MXDumpProfile + aggregate stats)."""
    if aggregate and _prof is not None:
        logging.info("\n%s", dumps())
    set_state("stop")


def memory_summary(device=None) -> str:
    """Device-memory profile (reference profiler memory profiling /
    gpu_memory_profiler): peak allocated/reserved plus the caching
    allocator's pool breakdown, from the substrate's accounting."""
    if not torch.cuda.is_available():
        return "(no GPU: memory profiling is device-side)"
    dev = device if device is not None else torch.cuda.current_device()
    alloc = torch.cuda.max_memory_allocated(dev) / 2 ** 30
    reserv = torch.cuda.max_memory_reserved(dev) / 2 ** 30
    head = (f"peak allocated {alloc:.2f} GiB, peak reserved {reserv:.2f} GiB "
            f"(of 288 GB HBM3E)\n")
    return head + torch.cuda.memory_summary(dev, abbreviated=True)
