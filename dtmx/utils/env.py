"""Typed env-var config tier (reference dmlc::GetEnv MXNET_* knobs,
docs/faq/env_var.md). Names kept where reference scripts
depend on them (ELASTIC_TRAINING_ENABLED, WORKER_HOST_FILE, NEW_WORKER,
EPOCH_BEGIN, DMLC_*, PS_*); dtmx-specific knobs use the DTMX_ prefix:

  DTMX_BUCKET_MB        all-reduce bucket size (default 50)
  DTMX_BACKEND          torch.distributed backend override (nccl|gloo)
  DTMX_STORE_TIMEOUT    rendezvous TCPStore timeout seconds (default 300)
"""
from __future__ import annotations

import os


def get_env_int(name: str, default: int) -> int:
    try:
        return int(os.environ.get(name, default))
    except ValueError:
        return default


def get_env_float(name: str, default: float) -> float:
    try:
        return float(os.environ.get(name, default))
    except ValueError:
        return default


def get_env_bool(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() in ("1", "true", "yes", "on")
