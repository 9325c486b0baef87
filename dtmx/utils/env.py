"""Typed env-var config tier (reference dmlc::GetEnv MXNET_* knobs,
docs/faq/env_var.md). Names kept where reference scripts
depend on them (ELASTIC_TRAINING_ENABLED, WORKER_HOST_FILE, NEW_WORKER,
EPOCH_BEGIN, DMLC_*); dtmx-specific knobs use the DTMX_ prefix.

Distributed / elastic:
  DTMX_BUCKET_MB        all-reduce bucket size MB (default 50; sized for the
                        7x ~153 GB/s xGMI links, not NVSwitch)
  DTMX_BACKEND          torch.distributed backend override (nccl|gloo)
  DTMX_STORE_TIMEOUT    rendezvous TCPStore timeout seconds (default 300)
  DTMX_PG_TIMEOUT       per-collective timeout seconds (bounded collectives;
                        mid-epoch peer death then fails loudly)
  PS_HEARTBEAT_INTERVAL / PS_HEARTBEAT_TIMEOUT
                        worker heartbeat cadence / staleness threshold
  PS_DROP_MSG           fault injection: probability of dropping a heartbeat
                        (reference van.cc:430 drop-message analog)
  DTMX_RUN_DIR          pidfile ledger dir for tools/kill_dtmx.py

Engine / debugging:
  DTMX_BLOCKING         1 = serialize kernel launches (NaiveEngine analog)
  DTMX_HIPGRAPH         1 = hipGraph step capture/replay in Module
  DTMX_FALLBACK         1 = route ops to the torch reference path
  DTMX_CHECK            1 = extra numerics checks in the op wrappers
  DTMX_PRINT_MEM        1 = print peak HBM use after bench
  DTMX_SEED / DTMX_TEST_SEED   RNG seeds (dtmx.random / tests)
  DTMX_BENCH_AUX        0 = skip bench.py's bs128 matched-batch aux row

Kernel dispatch (all measured defaults; the off-switches exist for A/B —
see profiles/resnet50_mi355x.md for the numbers behind each):
  DTMX_FUSED_BLOCK      0 = disable fused ResNet block backward
  DTMX_FUSE_BN_STATS    0 = standalone BN fwd stats (no GEMM epilogue fuse)
  DTMX_FUSE_BN_BWD      0 = standalone BN bwd stats (no EpiBnBwd epilogue)
  DTMX_FUSE_BN_CROSS    0 = disable cross-block BN bwd fusion handle
  DTMX_BN_DX_COEF       0 = direct-form bn_bwd_dx (no coefficient tables)
  DTMX_BN_COL           0 = strided-chunk BN elementwise kernels (no
                        fixed-column register-resident tables)
  DTMX_DISABLE_GEMM256  1 = never route to the 256^2 8-phase GEMM
  DTMX_NT_GROUP         0/1 = force XCD-grouped wgrad split-K off/on
                        (default: auto, only the tile class that wins)
  DTMX_WGRAD_TARGET_BLOCKS   wgrad split-K grid target (default 512)
  DTMX_NT_DEPTH         3 = triple-buffer NT LDS ring (measured worse)
  DTMX_NT_WM1NJ2        1 = 64x128 flat wgrad tile probe (measured worse)
  DTMX_G256_BAR1 / DTMX_G256_ST16 / DTMX_G256_M32
                        gemm256 probe variants (two-barrier phase, st_16x32
                        swizzle, 32x32x16 MFMA — all measured worse)
  DTMX_DISABLE_WGRAD_WS 1 = no persistent wgrad fp32 workspace
  DTMX_DISABLE_WT_KERNEL, DTMX_BK_ONLY_CONV, DTMX_TORCH_DROPOUT
                        narrower kernel-path off-switches
  DTMX_CUS / DTMX_XCDS / DTMX_WAVE   hardware-geometry overrides (tests)
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
