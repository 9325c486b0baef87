"""KVStore: the data-parallel engine (reference src/kvstore/*, python/mxnet/kvstore.py).

MI355X-native redesign: the reference's three-tier parameter
server (ps-lite workers/servers/scheduler, ZPush/ZPull per key) collapses into
**one elastic RCCL communicator group** over xGMI, one process per GPU:

  - `push(grads)` = all-reduce SUM over the group (RCCL ring over 7 xGMI
    links), `pull(weights)` = local read — every rank applies the *identical*
    optimizer update to its replica, which is exactly the reference's
    `update_on_kvstore` sync semantics (server merged grads until quorum ==
    NumWorkers, applied optimizer once: kvstore_dist_server.h:345-380).
  - aux keys ("exclude_update": BN running stats) are **averaged**, never
    optimizer-updated — the reference's key>=10M rule
    (kvstore_dist_server.h:353-360, kvstore_local.h:459).
  - the scheduler/Postoffice control plane becomes a small TCP rendezvous
    (dtmx.parallel.rendezvous) that versions the communicator; the
    membership-change barrier re-forms the group at epoch boundaries
    (reference MEMBERSHIP_CHANGE_BARRIER, ps-lite van.cc:256-292).

Backends: "nccl" (= RCCL on ROCm) for GPU groups, "gloo" for CPU test groups
(world_size > 1 works without GPUs — this is how tests/ exercise the
distributed path).
"""
from __future__ import annotations

import logging
import os
import pickle
from typing import Dict, List, Optional, Union

import torch
import torch.distributed as dist

from .optimizer import Optimizer, Updater, get_updater

# reference include/mxnet/kvstore.h:41 — keys >= this are "aux" keys that the
# server averages instead of applying the optimizer to.
MAX_ALLOWED_KEY_FOR_UPDATE = 10_000_000


def _as_list(x):
    return x if isinstance(x, (list, tuple)) else [x]


class KVStore:
    """Base interface (reference include/mxnet/kvstore.h / python kvstore.py)."""

    def __init__(self):
        self._updater: Optional[Updater] = None
        self._optimizer: Optional[Optimizer] = None
        # string->int key mapping with a separate counter for aux keys
        # (reference kvstore_local.h:459)
        self._str_key_dict: Dict[str, int] = {}
        self._next_key = 0
        self._next_aux_key = MAX_ALLOWED_KEY_FOR_UPDATE

    # -- key handling ------------------------------------------------------
    def _resolve_key(self, key: Union[int, str], exclude_update: bool = False) -> int:
        if isinstance(key, int):
            return key
        if key not in self._str_key_dict:
            if exclude_update:
                self._str_key_dict[key] = self._next_aux_key
                self._next_aux_key += 1
            else:
                self._str_key_dict[key] = self._next_key
                self._next_key += 1
        return self._str_key_dict[key]

    @staticmethod
    def _is_aux_key(ikey: int) -> bool:
        return ikey >= MAX_ALLOWED_KEY_FOR_UPDATE

    # -- interface ---------------------------------------------------------
    def init(self, key, value, exclude_update: bool = False):
        raise NotImplementedError

    def push(self, key, value, priority: int = 0):
        raise NotImplementedError

    def pull(self, key, out=None, priority: int = 0):
        raise NotImplementedError

    def set_optimizer(self, optimizer: Optimizer):
        self._optimizer = optimizer
        self._set_updater(get_updater(optimizer))

    def _set_updater(self, updater: Updater):
        self._updater = updater

    def set_gradient_compression(self, compression_params):
        raise NotImplementedError(f"{self.type} kvstore does not support compression")

    def row_sparse_pull(self, key, out=None, priority: int = 0, row_ids=None):
        """Pull selected rows of a 2-D value (reference kvstore.py:314 /
        kvstore_dist.h PullRowSparse_): each `row_ids` tensor selects rows
        of the stored value; unselected rows are zero in the target."""
        if row_ids is None:
            return self.pull(key, out=out, priority=priority)
        stored = self.pull(key)[0]
        rid_list = (list(row_ids) if isinstance(row_ids, (list, tuple))
                    else [row_ids])
        targets = (_as_list(out) if out is not None
                   else [torch.zeros_like(stored) for _ in rid_list])
        for dst, rids in zip(targets, rid_list):
            rids = rids.to(torch.long)
            dst.zero_()
            dst[rids] = stored[rids].to(dst.dtype)
        return None if out is not None else targets

    @property
    def type(self) -> str:
        raise NotImplementedError

    @property
    def rank(self) -> int:
        return 0

    @property
    def num_workers(self) -> int:
        return 1

    @property
    def num_dead_node(self) -> int:
        return 0

    def _barrier(self):
        pass

    def _membership_change_barrier(self, env: Dict[str, str]) -> bool:
        """Returns True if membership changed (group re-formed)."""
        return False

    def save_optimizer_states(self, fname: str, dump_optimizer: bool = False):
        if self._updater is None:
            raise RuntimeError("cannot save states without an optimizer set")
        payload = {
            "states": {
                k: _state_to_cpu(v) for k, v in self._updater.get_states().items()
            }
        }
        if dump_optimizer:
            payload["optimizer"] = self._optimizer
        with open(fname, "wb") as f:
            pickle.dump(payload, f)

    def load_optimizer_states(self, fname: str):
        if self._updater is None:
            raise RuntimeError("cannot load states without an optimizer set")
        with open(fname, "rb") as f:
            payload = pickle.load(f)
        self._updater.set_states(payload["states"])

    def close(self):
        pass


def _apply_row_sparse(store, ikey: int, urows: torch.Tensor,
                      merged: torch.Tensor, world: int):
    """Apply merged row-sparse values to a store entry: aux keys average the
    touched rows; with an optimizer, use its lazy row update when offered
    (reference row-sparse optimizer kernels, optimizer_op-inl.h:563-676 —
    untouched rows stay put); otherwise densify and run the full updater."""
    stored = store._store[ikey]
    if store._is_aux_key(ikey):
        stored[urows] = (merged / world).to(stored.dtype)
    elif store._updater is not None:
        opt = getattr(store._updater, "optimizer", None)
        if opt is not None and hasattr(opt, "update_rows"):
            st = store._updater.states.get(ikey)
            if st is None:
                st = opt.create_state(ikey, stored)
                store._updater.states[ikey] = st
            opt.update_rows(ikey, stored, urows, merged, st)
        else:
            dense = torch.zeros_like(stored, dtype=torch.float32)
            dense[urows] = merged
            store._updater(ikey, dense.to(stored.dtype), stored)
    else:
        stored[urows] = merged.to(stored.dtype)


def _state_to_cpu(state):
    if isinstance(state, torch.Tensor):
        return state.detach().cpu()
    if isinstance(state, tuple):
        return tuple(_state_to_cpu(s) for s in state)
    return state


class LocalKVStore(KVStore):
    """Single-process store (reference kvstore 'local'/'device',
    src/kvstore/kvstore_local.h). Multi-value pushes (one per device) are
    reduced by SUM on the first value's device — the CommDevice analog; on
    MI355X real multi-GPU runs use one process per GPU ('dist_*')."""

    def __init__(self, name: str = "local"):
        super().__init__()
        self._name = name
        self._store: Dict[int, torch.Tensor] = {}

    @property
    def type(self):
        return self._name

    def init(self, key, value, exclude_update: bool = False):
        keys = _as_list(key)
        values = _as_list(value)
        for k, v in zip(keys, values):
            ikey = self._resolve_key(k, exclude_update)
            self._store[ikey] = v.detach().clone()

    def _reduce(self, values: List[torch.Tensor]) -> torch.Tensor:
        merged = values[0].detach().clone()
        for v in values[1:]:
            merged += v.detach().to(merged.device)
        return merged

    def push(self, key, value, priority: int = 0):
        from .ops.functional import RowSparse

        keys = _as_list(key)
        if isinstance(value, RowSparse):
            ikey = self._resolve_key(keys[0])
            urows, inverse = torch.unique(value.rows.to(torch.long),
                                          sorted=True, return_inverse=True)
            merged = torch.zeros(urows.numel(), value.values.shape[1],
                                 dtype=torch.float32,
                                 device=value.values.device)
            merged.index_add_(0, inverse, value.values.float())
            _apply_row_sparse(self, ikey, urows, merged, 1)
            return
        for k, grouped in zip(keys, self._group_values(keys, value)):
            ikey = self._resolve_key(k)
            merged = self._reduce(grouped)
            stored = self._store[ikey]
            if self._is_aux_key(ikey):
                stored.copy_(merged / max(1, len(grouped)))
            elif self._updater is not None:
                self._updater(ikey, merged, stored)
            else:
                stored.copy_(merged)

    def pull(self, key, out=None, priority: int = 0):
        keys = _as_list(key)
        outs = self._group_values(keys, out)
        results = []
        for k, ogrp in zip(keys, outs):
            ikey = self._resolve_key(k)
            stored = self._store[ikey]
            if ogrp is None:
                results.append(stored)
            else:
                for o in ogrp:
                    o.detach().copy_(stored.to(o.device))
        return results if out is None else None

    @staticmethod
    def _group_values(keys, value):
        """Match the reference's grouping: one key may carry a list of
        per-device values."""
        if value is None:
            return [None] * len(keys)
        values = _as_list(value)
        if len(keys) == 1:
            if isinstance(value, (list, tuple)) and value and isinstance(value[0], (list, tuple)):
                return [list(value[0])]
            return [list(values)]
        grouped = []
        for v in values:
            grouped.append(list(_as_list(v)))
        return grouped


class DistKVStore(KVStore):
    """Distributed synchronous store over torch.distributed (RCCL on GPU,
    gloo on CPU). One process per worker; replaces the reference's
    KVStoreDist + KVStoreDistServer pair (src/kvstore/kvstore_dist.h,
    kvstore_dist_server.h) with collective all-reduce + replicated local
    update.
    """

    def __init__(self, name: str = "dist_sync"):
        super().__init__()
        self._name = name
        self._store: Dict[int, torch.Tensor] = {}
        self._compression = None
        from .parallel.rendezvous import ElasticContext

        self._elastic = ElasticContext.create_from_env()
        self._is_new_worker = os.environ.get("NEW_WORKER", "0") == "1"
        if not dist.is_initialized():
            self._elastic.init_group()

    @property
    def type(self):
        return self._name

    @property
    def rank(self):
        return dist.get_rank() if dist.is_initialized() else 0

    @property
    def num_workers(self):
        return dist.get_world_size() if dist.is_initialized() else 1

    @property
    def num_dead_node(self):
        return self._elastic.num_dead_node()

    @property
    def is_new_worker(self):
        return self._is_new_worker

    def init(self, key, value, exclude_update: bool = False):
        """Rank-0's value wins (reference: only rank 0 pushes init,
        kvstore_dist.h:205-224); new/recovering workers receive cluster state
        instead of pushing their own init."""
        keys = _as_list(key)
        values = _as_list(value)
        for k, v in zip(keys, values):
            ikey = self._resolve_key(k, exclude_update)
            t = v.detach().clone()
            dist.broadcast(t, src=0)
            self._store[ikey] = t
            if v.data_ptr() != t.data_ptr():
                v.detach().copy_(t)

    def push(self, key, value, priority: int = 0):
        from .ops.functional import RowSparse

        keys = _as_list(key)
        values = _as_list(value) if not isinstance(value, RowSparse) else [value]
        for k, v in zip(keys, values):
            ikey = self._resolve_key(k)
            if isinstance(v, RowSparse):
                self._push_row_sparse(ikey, v)
                continue
            if isinstance(v, (list, tuple)):  # per-device list: local reduce first
                merged = v[0].detach().clone()
                for extra in v[1:]:
                    merged += extra.detach().to(merged.device)
            else:
                merged = v.detach().clone()
            if self._compression is not None and not self._is_aux_key(ikey):
                merged = self._compression.compress_decompress(merged, key=ikey)
            dist.all_reduce(merged, op=dist.ReduceOp.SUM)
            stored = self._store[ikey]
            if self._is_aux_key(ikey):
                # aux keys: plain average (kvstore_dist_server.h:353-360)
                stored.copy_(merged / self.num_workers)
            elif self._updater is not None:
                self._updater(ikey, merged, stored)
            else:
                stored.copy_(merged)

    def pull(self, key, out=None, priority: int = 0):
        keys = _as_list(key)
        if out is None:
            return [self._store[self._resolve_key(k)] for k in keys]
        outs = _as_list(out)
        for k, o in zip(keys, outs):
            stored = self._store[self._resolve_key(k)]
            if isinstance(o, (list, tuple)):
                for oo in o:
                    oo.detach().copy_(stored.to(oo.device))
            else:
                o.detach().copy_(stored.to(o.device))
        return None

    def _push_row_sparse(self, ikey: int, v):
        """Row-sparse push: move only the touched rows over the wire and
        update only those rows (reference kvstore_dist.h:452-481
        PushRowSparse + kvstore_dist_server.h:503 DataHandleRowSparse;
        GPU row-merge replaces kvstore_utils.cu's cub unique). Wire format:
        length exchange -> padded all-gather of (rows, values) on the live
        backend (RCCL on GPU, gloo on CPU) -> local unique+segment-sum."""
        rows, vals = v.rows.to(torch.long), v.values.detach()
        world = self.num_workers
        if world > 1:
            dev = vals.device
            n = torch.tensor([rows.numel()], dtype=torch.long, device=dev)
            ns = [torch.zeros_like(n) for _ in range(world)]
            dist.all_gather(ns, n)
            counts = [int(x.item()) for x in ns]
            maxn = max(max(counts), 1)
            D = vals.shape[1]
            rbuf = torch.full((maxn,), -1, dtype=torch.long, device=dev)
            vbuf = torch.zeros(maxn, D, dtype=vals.dtype, device=dev)
            rbuf[: rows.numel()] = rows.to(dev)
            vbuf[: rows.numel()] = vals
            rgat = [torch.empty_like(rbuf) for _ in range(world)]
            vgat = [torch.empty_like(vbuf) for _ in range(world)]
            dist.all_gather(rgat, rbuf)
            dist.all_gather(vgat, vbuf)
            all_rows = torch.cat([r[:c] for r, c in zip(rgat, counts)])
            all_vals = torch.cat([vv[:c] for vv, c in zip(vgat, counts)])
        else:
            all_rows, all_vals = rows, vals
        urows, inverse = torch.unique(all_rows, sorted=True,
                                      return_inverse=True)
        merged = torch.zeros(urows.numel(), all_vals.shape[1],
                             dtype=torch.float32, device=all_vals.device)
        merged.index_add_(0, inverse, all_vals.float())
        _apply_row_sparse(self, ikey, urows, merged, world)

    def set_gradient_compression(self, compression_params):
        from .parallel.compression import TwoBitCompression

        params = dict(compression_params)
        ctype = params.get("type", "2bit")
        if ctype == "none":
            self._compression = None
        elif ctype == "2bit":
            self._compression = TwoBitCompression(params.get("threshold", 0.5))
        else:
            raise ValueError(f"unknown compression type {ctype}")

    def _barrier(self):
        dist.barrier()

    def _membership_change_barrier(self, env: Dict[str, str]) -> bool:
        """Epoch-boundary membership barrier (reference kvstore_dist.h:117-122
        -> postoffice.cc:329-381 -> ETNodeManager). All current workers enter;
        the rendezvous diffs the roster; on change the communicator is
        destroyed and re-formed with dense re-ranking. Returns True when the
        worker set changed. Workers slated for removal do NOT return — they
        exit the process cleanly after the handoff."""
        return self._elastic.membership_change_barrier(env)

    def broadcast_state(self, tensors: List[torch.Tensor], src: int = 0):
        """Bulk state sync for joiners (reference: joiner pulls weights+aux
        from servers, model.py:116-133)."""
        for t in tensors:
            dist.broadcast(t.detach(), src=src)

    def close(self):
        self._elastic.shutdown()


def create(name: str = "local") -> KVStore:
    """Factory (reference kvstore.cc:40-75 / python kvstore.py create)."""
    name = name.lower()
    if name in ("local", "local_update_cpu", "local_allreduce_cpu", "device", "nccl_local"):
        return LocalKVStore(name)
    if name in ("dist_sync", "dist_device_sync", "dist_async", "dist_sync_device", "nccl", "dist"):
        # single-process fallback: behave like local when no dist env is set
        dist_env = any(
            k in os.environ
            for k in ("RANK", "WORLD_SIZE", "DMLC_ROLE", "DMLC_PS_ROOT_URI", "DMLC_WORKER_ID")
        ) or os.environ.get("ELASTIC_TRAINING_ENABLED", "0").lower() in ("1", "true")
        if not dist_env:
            logging.warning("kvstore '%s' requested without a distributed launcher; using local", name)
            return LocalKVStore(name)
        return DistKVStore(name)
    raise ValueError(f"unknown kvstore type {name}")
