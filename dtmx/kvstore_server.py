"""Server-role compat shim (reference python/mxnet/kvstore_server.py).

dtmx has no separate parameter-server tier: sync DP runs as collective
all-reduce with replicated local updates, so a process
launched with DMLC_ROLE=server has nothing to serve. The launcher
(tools/launch.py) does not spawn servers; this module exists so reference
scripts that import it keep working, and to document the mapping.
"""
from __future__ import annotations

import logging
import os


def _init_kvstore_server_module():
    role = os.environ.get("DMLC_ROLE", "worker")
    if role == "server":
        logging.info(
            "dtmx: DMLC_ROLE=server is a no-op (no PS tier; sync DP is an "
            "elastic RCCL all-reduce group). Exiting cleanly."
        )
        raise SystemExit(0)
    if role == "scheduler":
        # the scheduler role is served by dtmx.parallel.rendezvous.Scheduler
        # inside tools/launch.py
        logging.info("dtmx: scheduler role is hosted by tools/launch.py")
