"""Elastic rendezvous: the MI355X-native replacement for ps-lite's
scheduler/Postoffice/Van control plane (reference 3rdparty/ps-lite/src/
postoffice.cc, van.cc, elastic_training.cc).

Design: a single TCP key-value store (torch TCPStore) hosted
by the launcher ("scheduler") versions the communicator. Each generation v
has a member list; workers form a torch.distributed process group per
generation with dense ranks = index in the sorted member list (the
reference's dense re-ranking, van.cc:519-539). Membership changes commit only
at epoch barriers (van.cc:256-292): every current worker enters
`membership_change_barrier`, rank 0 reads the published roster version, and
on a change the group is destroyed and re-formed. Joiners (NEW_WORKER=1) are
launched with the *new* generation already published, so their initial
`init_group` completes exactly when the survivors re-form — the reference's
"node count target already raised" join protocol. Removed
workers exit after the barrier; their ranks disappear densely.

Scheduler-side rules preserved from the reference (elastic_training.cc):
  - removal beats addition within one diff (:66-77): the scheduler publishes
    the removal generation first, the addition as the next generation;
  - initial workers are not removable (README.md:54-59);
  - an append-only membership audit log `<hostfile>_log` with
    "SEQNUM ADDED|REMOVED id time" lines (:108-126).

Environment contract (names kept where reference scripts depend on them —
SURVEY.md §5.6):
  DMLC_PS_ROOT_URI / DMLC_PS_ROOT_PORT   rendezvous host/port (elastic mode)
  DMLC_NUM_WORKER                        initial world size (elastic mode)
  DMLC_WORKER_ID                         stable worker identity
  NEW_WORKER=1, EPOCH_BEGIN=<n>          joiner bootstrap
  ELASTIC_TRAINING_ENABLED=1             enable elastic path
  WORKER_HOST_FILE                       roster file (read by the scheduler)
  RANK/WORLD_SIZE/MASTER_ADDR/...        static (torchrun) mode
"""
from __future__ import annotations

import datetime
import json
import logging
import os
import socket
import sys
import threading
import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

log = logging.getLogger("dtmx.elastic")

_HB_INTERVAL = float(os.environ.get("PS_HEARTBEAT_INTERVAL", "5"))
_HB_TIMEOUT = float(os.environ.get("PS_HEARTBEAT_TIMEOUT", "30"))


def _backend() -> str:
    b = os.environ.get("DTMX_BACKEND")
    if b:
        return b
    return "nccl" if torch.cuda.is_available() else "gloo"


def _store_timeout() -> datetime.timedelta:
    return datetime.timedelta(seconds=float(os.environ.get("DTMX_STORE_TIMEOUT", "300")))


class Scheduler:
    """Roster authority. Runs inside the launcher (tools/launch.py) or a test
    harness; hosts the TCPStore master and publishes roster generations."""

    def __init__(self, host: str, port: int, initial_members: List[str],
                 hostfile: Optional[str] = None):
        self.store = dist.TCPStore(host, port, is_master=True,
                                   timeout=_store_timeout(),
                                   wait_for_workers=False)
        self.initial_members = sorted(initial_members)
        self.members = list(self.initial_members)
        self.version = 1
        self.hostfile = hostfile
        self.seqnum = 0
        self.store.set(f"roster/members/{self.version}", json.dumps(self.members))
        self.store.add("roster/version", 1)
        for m in self.members:
            self.store.set(f"hb/{m}", str(time.time()))

    # -- roster mutation ---------------------------------------------------
    def publish(self, new_members: List[str]) -> List[List[str]]:
        """Publish a roster change; returns the list of generations created.
        Removal beats addition: a mixed diff becomes two generations."""
        new_members = sorted(set(new_members))
        cur = set(self.members)
        tgt = set(new_members)
        removed = cur - tgt
        added = tgt - cur
        # initial workers are not removable (reference README.md:54-59)
        blocked = removed & set(self.initial_members)
        if blocked:
            log.warning("refusing to remove initial workers: %s", sorted(blocked))
            removed -= blocked
        generations = []
        if removed:
            step = sorted(cur - removed)
            self._emit(step, removed=sorted(removed))
            generations.append(step)
        if added:
            step = sorted(set(self.members) | added)
            self._emit(step, added=sorted(added))
            generations.append(step)
        return generations

    def _emit(self, members: List[str], added=None, removed=None):
        self.version += 1
        self.members = members
        self.store.set(f"roster/members/{self.version}", json.dumps(members))
        for m in added or []:
            self.store.set(f"hb/{m}", str(time.time()))
        self.store.add("roster/version", 1)
        now = time.time()
        for m in added or []:
            self._log_line("ADDED", m, now)
        for m in removed or []:
            self._log_line("REMOVED", m, now)
        log.info("published roster v%d: %s", self.version, members)

    def _log_line(self, what: str, member: str, t: float):
        # audit log format of elastic_training.cc:108-126
        self.seqnum += 1
        if self.hostfile:
            with open(self.hostfile + "_log", "a") as f:
                f.write(f"{self.seqnum} {what} {member} {t}\n")

    def watch_hostfile(self, poll_seconds: float = 1.0, stop_event: Optional[threading.Event] = None):
        """Poll WORKER_HOST_FILE and publish diffs (reference
        findMembershipChanges, elastic_training.cc:135-157)."""
        assert self.hostfile
        last: Optional[List[str]] = None
        while stop_event is None or not stop_event.is_set():
            try:
                members = read_hostfile(self.hostfile)
            except FileNotFoundError:
                members = None
            if members is not None and members != last:
                if last is not None:
                    self.publish(members)
                last = members
            time.sleep(poll_seconds)

    def num_dead(self) -> int:
        return len(self.dead_members())

    def dead_members(self) -> List[str]:
        out = []
        now = time.time()
        for m in self.members:
            try:
                ts = float(self.store.get(f"hb/{m}"))
                if now - ts > _HB_TIMEOUT:
                    out.append(m)
            except Exception:
                out.append(m)
        return out

    def prune_dead(self) -> List[str]:
        """Publish a roster without heartbeat-expired members (UNPLANNED
        failure handling; reference postoffice.cc:410-429 GetDeadNodes +
        van.cc dead-node accounting). Unlike planned removal, death pruning
        may remove initial workers — a dead node cannot object — so this
        emits directly rather than via publish()'s policy filter. Survivors
        blocked on a timed-out collective pick up the new generation at
        their next barrier (or exit loudly mid-epoch)."""
        dead = self.dead_members()
        if not dead or len(dead) >= len(self.members):
            return []
        survivors = sorted(set(self.members) - set(dead))
        self._emit(survivors, removed=sorted(dead))
        return dead


def read_hostfile(path: str) -> List[str]:
    """Host file -> worker ids. Duplicate host lines (e.g. the reference's
    tools/host_worker with two 127.0.0.1 lines) get per-occurrence suffixes
    so localhost simulation works."""
    seen: Dict[str, int] = {}
    ids = []
    with open(path) as f:
        for line in f:
            h = line.strip()
            if not h or h.startswith("#"):
                continue
            k = seen.get(h, 0)
            seen[h] = k + 1
            ids.append(f"{h}#{k}")
    return ids


class ElasticContext:
    """Per-worker communicator manager."""

    def __init__(self, mode: str, worker_id: str = "", uri: str = "", port: int = 0):
        self.mode = mode  # "static" | "elastic" | "single"
        self.worker_id = worker_id
        self.uri = uri
        self.port = port
        self.store: Optional[dist.TCPStore] = None
        self.version = 0
        self.members: List[str] = []
        self._hb_stop: Optional[threading.Event] = None

    # -- construction ------------------------------------------------------
    @staticmethod
    def create_from_env() -> "ElasticContext":
        elastic = os.environ.get("ELASTIC_TRAINING_ENABLED", "0").lower() in ("1", "true")
        if elastic or "DMLC_PS_ROOT_URI" in os.environ:
            uri = os.environ.get("DMLC_PS_ROOT_URI", "127.0.0.1")
            port = int(os.environ.get("DMLC_PS_ROOT_PORT", "9091"))
            wid = os.environ.get("DMLC_WORKER_ID") or f"{socket.gethostname()}#{os.getpid()}"
            return ElasticContext("elastic", wid, uri, port)
        if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
            return ElasticContext("static")
        return ElasticContext("single")

    # -- group lifecycle ---------------------------------------------------
    def init_group(self):
        if self.mode == "single":
            return
        if self.mode == "static":
            if not dist.is_initialized():
                dist.init_process_group(backend=_backend())
            return
        # elastic
        self.store = dist.TCPStore(self.uri, self.port, is_master=False,
                                   timeout=_store_timeout())
        self.store.set(f"hb/{self.worker_id}", str(time.time()))
        self._start_heartbeat()
        version = self.store.add("roster/version", 0)
        members = self._read_members(version)
        # A joiner may start before the scheduler published the generation
        # that includes it; wait for a roster containing us.
        while self.worker_id not in members:
            time.sleep(0.2)
            version = self.store.add("roster/version", 0)
            members = self._read_members(version)
        self._form(version, members)

    def _read_members(self, version: int) -> List[str]:
        return json.loads(self.store.get(f"roster/members/{version}").decode())

    def _form(self, version: int, members: List[str]):
        rank = members.index(self.worker_id)
        world = len(members)
        prefix = dist.PrefixStore(f"gen{version}", self.store)
        # bounded collectives: when a peer dies MID-epoch (no barrier in
        # sight), the next all-reduce must fail loudly after DTMX_PG_TIMEOUT
        # instead of hanging the survivors forever (reference analog: ps-lite
        # heartbeat timeout surfacing as van.cc dead-node handling)
        timeout = datetime.timedelta(
            seconds=float(os.environ.get("DTMX_PG_TIMEOUT", "300")))
        dist.init_process_group(backend=_backend(), store=prefix,
                                rank=rank, world_size=world, timeout=timeout)
        self.version = version
        self.members = members
        if torch.cuda.is_available():
            # one process per GPU; LOCAL_RANK when the launcher pins it
            local = int(os.environ.get("LOCAL_RANK", rank % max(1, torch.cuda.device_count())))
            torch.cuda.set_device(local)
        log.info("joined generation %d as rank %d/%d", version, rank, world)

    # -- membership barrier -------------------------------------------------
    def membership_change_barrier(self, env: Dict[str, str]) -> bool:
        if self.mode != "elastic" or self.store is None:
            if dist.is_initialized():
                dist.barrier()
            return False
        dist.barrier()
        # rank 0 reads the roster version; everyone agrees via broadcast
        decision = [0, None]
        if dist.get_rank() == 0:
            # publish the barrier's env payload (EPOCH_BEGIN etc.) so the
            # launcher can hand it to ssh-launched joiners (reference
            # ETNodeManager launchCommandOnNewWorker, elastic_training.cc:26-62)
            for ek, ev in (env or {}).items():
                self.store.set(f"cluster/env/{ek}", str(ev))
            if env and "EPOCH_BEGIN" in env:
                self.store.set("cluster/epoch", str(env["EPOCH_BEGIN"]))
            v = self.store.add("roster/version", 0)
            decision = [v, self._read_members(v) if v != self.version else None]
        obj = [decision]
        dist.broadcast_object_list(obj, src=0)
        version, members = obj[0]
        if members is None or version == self.version:
            return False
        # re-form; quiesce in-flight RCCL work first — destroying the NCCL
        # communicator with collectives still on-stream is undefined
        # (reference analog: engine WaitForAll before group teardown)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dist.destroy_process_group()
        if self.worker_id not in members:
            log.info("worker %s removed from roster at v%d; exiting", self.worker_id, version)
            self._stop_heartbeat()
            sys.exit(0)
        self._form(version, members)
        return True

    def num_dead_node(self) -> int:
        if self.store is None:
            return 0
        n = 0
        now = time.time()
        for m in self.members:
            try:
                ts = float(self.store.get(f"hb/{m}"))
                if now - ts > _HB_TIMEOUT:
                    n += 1
            except Exception:
                n += 1
        return n

    # -- heartbeats (reference van.cc:686-698) ------------------------------
    def _start_heartbeat(self):
        self._hb_stop = threading.Event()
        # PS_DROP_MSG fault injection (reference ps-lite van.cc:430-432:
        # random message drop to exercise failure detection): probability of
        # dropping each heartbeat. PS_DROP_MSG=1 makes this worker appear
        # dead after PS_HEARTBEAT_TIMEOUT without killing it.
        drop_p = float(os.environ.get("PS_DROP_MSG", "0"))

        def beat():
            import random

            while not self._hb_stop.wait(_HB_INTERVAL):
                if drop_p > 0 and random.random() < drop_p:
                    continue
                try:
                    self.store.set(f"hb/{self.worker_id}", str(time.time()))
                except Exception:
                    return

        threading.Thread(target=beat, daemon=True).start()

    def _stop_heartbeat(self):
        if self._hb_stop is not None:
            self._hb_stop.set()

    def shutdown(self):
        self._stop_heartbeat()
        if dist.is_initialized():
            dist.destroy_process_group()
