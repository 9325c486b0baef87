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


def dump():
    set_state("stop")
