"""Checkpoint helpers (reference python/mxnet/model.py:434-492).

`save_checkpoint(prefix, epoch, ...)` writes
  <prefix>-symbol.json     : the model spec (dtmx models serialize their
                             constructor spec; reference saved the NNVM graph)
  <prefix>-%04d.params     : single V2-NDArray file of
                             {"arg:<name>"|"aux:<name>" -> tensor}
with the params file byte-compatible with the reference format so reference
tooling can read dtmx checkpoints.
"""
from __future__ import annotations

import json
import logging
from typing import Dict, Optional, Tuple

import torch

from . import ndarray as nd


def save_checkpoint(prefix: str, epoch: int, symbol, arg_params: Dict[str, torch.Tensor],
                    aux_params: Dict[str, torch.Tensor]):
    if symbol is not None:
        with open(f"{prefix}-symbol.json", "w") as f:
            json.dump(_symbol_to_json(symbol), f, indent=2)
    save_dict = {f"arg:{k}": v for k, v in arg_params.items()}
    save_dict.update({f"aux:{k}": v for k, v in aux_params.items()})
    param_name = "%s-%04d.params" % (prefix, epoch)
    nd.save(param_name, save_dict)
    logging.info('Saved checkpoint to "%s"', param_name)


def load_checkpoint(prefix: str, epoch: int):
    symbol = None
    try:
        with open(f"{prefix}-symbol.json") as f:
            symbol = json.load(f)
    except FileNotFoundError:
        pass
    loaded = nd.load("%s-%04d.params" % (prefix, epoch))
    arg_params, aux_params = {}, {}
    for k, v in loaded.items():
        tp, name = k.split(":", 1)
        if tp == "arg":
            arg_params[name] = v
        elif tp == "aux":
            aux_params[name] = v
    return symbol, arg_params, aux_params


def _symbol_to_json(symbol):
    spec = getattr(symbol, "spec", None)
    if spec is not None:
        return spec
    if isinstance(symbol, dict):
        return symbol
    return {"repr": repr(symbol)}


def _create_kvstore(kvstore, num_device: int, arg_params):
    """Reference model.py:77-113: resolve a kvstore spec to (kv,
    update_on_kvstore)."""
    from . import kvstore as kvs

    update_on_kvstore = True
    if kvstore is None:
        kv = None
    elif isinstance(kvstore, kvs.KVStore):
        kv = kvstore
    elif isinstance(kvstore, str):
        if num_device == 1 and "dist" not in kvstore:
            kv = kvs.create(kvstore)
        else:
            kv = kvs.create(kvstore)
            if kvstore == "local":
                max_size = max(p.numel() for p in arg_params.values()) if arg_params else 0
                if max_size > 1024 * 1024:
                    update_on_kvstore = False
    else:
        raise TypeError("kvstore must be KVStore, str or None")
    return kv, update_on_kvstore
