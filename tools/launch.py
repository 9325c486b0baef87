#!/usr/bin/env python3
"""Elastic cluster launcher (reference tools/launch.py + dmlc_tracker ssh).

Launches N workers (locally or over ssh), hosts the rendezvous scheduler
(the ps-lite scheduler/Postoffice replacement, dtmx.parallel.rendezvous),
watches the worker host file, and launches joiners when hosts are added —
the reference's ETNodeManager/EC2 control loop collapsed into one process.

    python tools/launch.py -n 2 -H hostfile --elastic-training-enabled True \
        python train.py --network resnet --num-layers 50 ...

Flags kept from the reference surface (tools/launch.py:40-85):
  -n/--num-workers, -H/--hostfile, --elastic-training-enabled,
  --launch-worker (single-worker mode used for joiners), --sync-dst-dir.

Mechanics:
  - hostfile lines are hosts (duplicates allowed — localhost simulation,
    reference tools/host_worker); worker id = "host#occurrence".
  - initial workers get DMLC_WORKER_ID/DMLC_PS_ROOT_URI/PORT/
    ELASTIC_TRAINING_ENABLED env; LOCAL_RANK = per-host index (GPU pinning).
  - on hostfile additions the scheduler publishes a new generation and this
    launcher starts the joiner with NEW_WORKER=1 and EPOCH_BEGIN read from
    the cluster env payload (reference elastic_training.cc:26-62).
  - removals publish a generation without the host; the worker exits itself
    at the next epoch barrier (membership audit log: <hostfile>_log).
  - initial workers are not removable; removal beats addition.
"""
from __future__ import annotations

import argparse
import logging
import os
import shlex
import signal
import subprocess
import sys
import threading
import time
from collections import defaultdict
from typing import Dict, List

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dtmx.parallel.rendezvous import Scheduler, read_hostfile  # noqa: E402


def is_local(host: str) -> bool:
    return host in ("127.0.0.1", "localhost", os.uname().nodename)


class Launcher:
    def __init__(self, args, command: List[str]):
        self.args = args
        self.command = command
        self.port = args.scheduler_port
        self.uri = args.scheduler_host
        self.procs: Dict[str, subprocess.Popen] = {}
        if args.hostfile:
            members = read_hostfile(args.hostfile)[: args.num_workers or None]
            if args.num_workers and len(members) < args.num_workers:
                raise SystemExit("hostfile has fewer hosts than -n")
        else:
            members = [f"127.0.0.1#{i}" for i in range(args.num_workers)]
        self.initial_members = members
        self.scheduler = Scheduler(self.uri, self.port, members,
                                   hostfile=args.hostfile)
        self.known = set(members)

    def worker_env(self, wid: str, new_worker: bool = False) -> Dict[str, str]:
        host = wid.split("#")[0]
        local_idx = len([w for w in self.procs if w.split("#")[0] == host])
        env = dict(os.environ)
        env.update(
            DMLC_WORKER_ID=wid,
            DMLC_PS_ROOT_URI=self.uri,
            DMLC_PS_ROOT_PORT=str(self.port),
            DMLC_NUM_WORKER=str(len(self.scheduler.members)),
            DMLC_ROLE="worker",
            ELASTIC_TRAINING_ENABLED="1",
            LOCAL_RANK=str(local_idx),
        )
        if new_worker:
            env["NEW_WORKER"] = "1"
            # EPOCH_BEGIN is only a hint (the joiner adopts the authoritative
            # cluster epoch at its first barrier) — bound the fetch: a blocking
            # store.get would stall 300 s when the cluster has not reached its
            # first epoch barrier yet.
            from datetime import timedelta
            try:
                self.scheduler.store.set_timeout(timedelta(seconds=2))
                env["EPOCH_BEGIN"] = self.scheduler.store.get("cluster/env/EPOCH_BEGIN").decode()
            except Exception:
                env["EPOCH_BEGIN"] = "0"
            finally:
                self.scheduler.store.set_timeout(timedelta(seconds=300))
        return env

    def prepare_data(self, wid: str) -> bool:
        """Admission gate for joiners (reference prepare-data.py workflow:
        the EC2 manager admits a host only after its data-prep success
        marker appears, README.md:100-110). Runs the configured script on
        the joiner's host; True = admit."""
        script = getattr(self.args, "prepare_data_script", None)
        if not script:
            return True
        host = wid.split("#")[0]
        if is_local(host):
            cmd = [script, wid]
        else:
            cmd = ["ssh", "-o", "StrictHostKeyChecking=no", host,
                   f"{shlex.quote(script)} {shlex.quote(wid)}"]
        r = subprocess.run(cmd)
        return r.returncode == 0

    def launch_worker(self, wid: str, new_worker: bool = False):
        host = wid.split("#")[0]
        env = self.worker_env(wid, new_worker)
        if is_local(host):
            p = subprocess.Popen(self.command, env=env)
        else:
            # ssh launch (reference dmlc_tracker ssh.submit)
            envstr = " ".join(
                f"{k}={shlex.quote(v)}"
                for k, v in env.items()
                if k.startswith(("DMLC_", "NEW_WORKER", "EPOCH_BEGIN", "ELASTIC", "LOCAL_RANK", "DTMX_"))
            )
            cmd = f"cd {shlex.quote(os.getcwd())} && env {envstr} " + " ".join(
                shlex.quote(c) for c in self.command
            )
            p = subprocess.Popen(["ssh", "-o", "StrictHostKeyChecking=no", host, cmd])
        self.procs[wid] = p
        # exact-PID ledger for tools/kill_dtmx.py (reference kill-mxnet.py
        # redesigned: pidfiles instead of process-name pattern kills)
        rundir = os.path.expanduser(os.environ.get("DTMX_RUN_DIR", "~/.dtmx/run"))
        try:
            os.makedirs(rundir, exist_ok=True)
            with open(os.path.join(rundir, wid.replace("/", "_") + ".pid"), "w") as f:
                f.write(str(p.pid))
        except OSError:
            pass
        logging.info("launched worker %s (pid %d, new=%s)", wid, p.pid, new_worker)

    def watch(self, stop: threading.Event):
        """Hostfile watcher -> roster publish -> joiner launches."""
        last = list(self.initial_members)
        while not stop.is_set():
            time.sleep(self.args.poll_seconds)
            # unplanned-death pruning: heartbeat-expired members leave the
            # roster so survivors re-form at their next barrier
            # (reference postoffice.cc:410-429 dead-node accounting)
            pruned = self.scheduler.prune_dead()
            for wid in pruned:
                logging.warning("pruned dead worker %s from roster", wid)
            if not self.args.hostfile:
                continue
            try:
                members = read_hostfile(self.args.hostfile)
            except FileNotFoundError:
                continue
            if members == last:
                continue
            before = set(self.scheduler.members)
            # admission gate BEFORE publish: a joiner that enters the roster
            # but never starts would strand the survivors' re-form (the
            # reference's marker gates hostfile entry the same way)
            admitted = [m for m in members
                        if m in before or self.prepare_data(m)]
            for m in set(members) - set(admitted):
                logging.error("prepare-data failed for %s; joiner NOT "
                              "admitted to the roster", m)
            self.scheduler.publish(admitted)
            added = set(self.scheduler.members) - before
            for wid in sorted(added):
                self.launch_worker(wid, new_worker=True)
            last = members

    def run(self) -> int:
        for wid in self.initial_members:
            self.launch_worker(wid)
        stop = threading.Event()
        t = threading.Thread(target=self.watch, args=(stop,), daemon=True)
        if self.args.elastic_training_enabled:
            t.start()
        rc = 0
        try:
            while True:
                live = {w: p for w, p in self.procs.items() if p.poll() is None}
                for w, p in self.procs.items():
                    if p.poll() not in (None, 0):
                        logging.error("worker %s exited rc=%d", w, p.returncode)
                        rc = p.returncode
                if not live:
                    break
                time.sleep(0.5)
        except KeyboardInterrupt:
            for p in self.procs.values():
                p.send_signal(signal.SIGINT)
        stop.set()
        return rc


def main():
    logging.basicConfig(level=logging.INFO)
    ap = argparse.ArgumentParser(description=__doc__,
                                 formatter_class=argparse.RawDescriptionHelpFormatter)
    ap.add_argument("-n", "--num-workers", type=int, default=0)
    ap.add_argument("-H", "--hostfile", type=str, default=None,
                    help="worker host file (WORKER_HOST_FILE); rewritten lines drive elasticity")
    ap.add_argument("--elastic-training-enabled", type=lambda s: s.lower() in ("1", "true"),
                    default=False)
    ap.add_argument("--launch-worker", type=lambda s: s.lower() in ("1", "true"),
                    default=False, help="launch a single (joining) worker and exit")
    ap.add_argument("--scheduler-host", type=str,
                    default=os.environ.get("DMLC_PS_ROOT_URI", "127.0.0.1"))
    ap.add_argument("--scheduler-port", type=int,
                    default=int(os.environ.get("DMLC_PS_ROOT_PORT", "9091")))
    ap.add_argument("--poll-seconds", type=float, default=1.0)
    ap.add_argument("--sync-dst-dir", type=str, default=None,
                    help="rsync working dir to remote hosts before launch")
    ap.add_argument("--prepare-data-script", type=str, default=None,
                    help="per-joiner data-preparation hook: run (locally or "
                         "over ssh) before a NEW worker launches; a nonzero "
                         "exit blocks admission (reference prepare-data.py "
                         "+ the prepare_data_success_<ip> EFS marker gate)")
    ap.add_argument("command", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    command = [c for c in args.command if c != "--"]
    if not command:
        ap.error("no training command given")
    if args.hostfile:
        os.environ["WORKER_HOST_FILE"] = args.hostfile

    if args.launch_worker:
        # single-worker joiner mode (reference launch.py:309-312): env is
        # expected to be pre-set by the caller (scheduler/launcher).
        os.execvpe(command[0], command, dict(os.environ, NEW_WORKER="1"))

    sys.exit(Launcher(args, command).run())


if __name__ == "__main__":
    main()
