"""Module: the elastic training loop (reference python/mxnet/module/
{base_module.py,module.py}).

dtmx's Module drives a torch nn.Module (built by dtmx.models) instead of an
NNVM symbol, but keeps the reference's training-loop contract:

  fit() epoch loop (base_module.py:412,503-605):
    - membership-change barrier at every epoch top (":540-543")
    - data iterator re-created when the worker count changes (":545-549")
    - forward_backward -> update -> update_metric per batch
    - store_aux_params (BN running stats, cluster-averaged) at epoch end
    - checkpoint via epoch_end_callback

  elastic joiner protocol (base_module.py:503-513, module.py:553-565):
    - begin_epoch = EPOCH_BEGIN env for NEW_WORKER processes
    - init_optimizer(initialize_from_kvstore=True): adopt cluster state
      (weights + aux + optimizer state) instead of pushing local init
    - the joiner skips the first membership barrier (it completed the roster)

  dynamic-minibatch SGD (the paper's recipe; fixes the reference's stale
  rescale, module.py:515-518): rescale_grad = 1/(batch_size*num_workers) is
  recomputed whenever the membership changes, and an attached
  WarmupScheduler is re-armed on worker joins.
"""
from __future__ import annotations

import logging
import os
import time
from typing import Callable, Dict, List, Optional, Sequence, Tuple, Union

import torch
import torch.distributed as dist

from .. import initializer as init_mod
from .. import metric as metric_mod
from .. import model as model_mod
from ..callback import BatchEndParam
from ..context import Context, cpu
from ..io import DataBatch, DataIter
from ..kvstore import KVStore, DistKVStore, LocalKVStore
from ..lr_scheduler import WarmupScheduler
from ..optimizer import Optimizer, Updater, create as opt_create, get_updater
from ..parallel.bucketer import GradBucketer


def _as_list(x):
    return x if isinstance(x, (list, tuple)) else [x]


class Module:
    def __init__(self, symbol, data_names=("data",), label_names=("softmax_label",),
                 logger=logging, context: Union[Context, Sequence[Context]] = None,
                 work_load_list=None, fixed_param_names=None):
        self.symbol = symbol  # an nn.Module (dtmx.models builder output)
        if not isinstance(symbol, torch.nn.Module):
            raise TypeError("dtmx Module drives a torch nn.Module (see dtmx.models)")
        self.data_names = list(data_names)
        self.label_names = list(label_names or [])
        self.logger = logger
        ctxs = _as_list(context) if context is not None else [cpu()]
        if len(ctxs) > 1:
            logger.warning(
                "dtmx runs one process per GPU (torchrun); using first of %s", ctxs
            )
        self.ctx: Context = ctxs[0]
        self.device = self.ctx.torch_device()
        self.fixed_param_names = set(fixed_param_names or [])

        self.binded = False
        self.params_initialized = False
        self.optimizer_initialized = False
        self.for_training = False
        self._dtype = torch.float32
        self._channels_last = False
        self._loss_scale = 1.0

        self._kvstore: Optional[KVStore] = None
        self._updater: Optional[Updater] = None
        self._optimizer: Optional[Optimizer] = None
        self._update_on_kvstore = True
        self._bucketer: Optional[GradBucketer] = None
        self._use_fused_sgd = False
        self._batch_size = 0
        # hipGraph step capture (HIP graphs in place of the reference's
        # op-segment bulking, graph_executor.cc:1318)
        self._capture_mode = False
        self._graph = None
        self._static_inputs = None
        self._hyper_dev = None

        self._outputs: List[torch.Tensor] = []
        self._loss: Optional[torch.Tensor] = None
        self._labels: List[torch.Tensor] = []

        # elastic data-iterator factory (reference BaseDataIterator handle,
        # module.py:75,111,198-202)
        self._data_iterator: Optional[Callable] = None

    # ------------------------------------------------------------------ bind
    def bind(self, data_shapes, label_shapes=None, for_training=True,
             inputs_need_grad=False, force_rebind=False, shared_module=None,
             grad_req="write", dtype: torch.dtype = None, channels_last: bool = None):
        if self.binded and not force_rebind:
            return
        self.for_training = for_training
        if dtype is not None:
            self._dtype = dtype
        if channels_last is None:
            channels_last = (
                self._dtype in (torch.bfloat16, torch.float16)
                and len(data_shapes[0][1]) == 4
            )
        self._channels_last = channels_last
        self.symbol.to(self.device)
        if self._dtype != torch.float32:
            self.symbol.to(self._dtype)
            # keep BN stats + small norm params in fp32 for stability:
            # dtmx BN layers handle internal fp32; torch BN buffers follow dtype
        if self._channels_last:
            self.symbol.to(memory_format=torch.channels_last)
        self._batch_size = data_shapes[0][1][0]
        self.data_shapes = data_shapes
        self.label_shapes = label_shapes
        self.binded = True
        if not for_training:
            self.symbol.eval()
        else:
            self.symbol.train()

    # ---------------------------------------------------------------- params
    def _classified_named_tensors(self):
        """arg (trainable) / aux (BN running stats) split, sorted by name for
        deterministic cross-rank iteration order."""
        args = dict(self.symbol.named_parameters())
        auxs = {
            k: v
            for k, v in self.symbol.named_buffers()
            if k.endswith("running_mean") or k.endswith("running_var")
        }
        return dict(sorted(args.items())), dict(sorted(auxs.items()))

    def get_params(self) -> Tuple[Dict[str, torch.Tensor], Dict[str, torch.Tensor]]:
        args, auxs = self._classified_named_tensors()
        to_cpu = lambda d: {k: v.detach().float().cpu() for k, v in d.items()}
        return to_cpu(args), to_cpu(auxs)

    def set_params(self, arg_params, aux_params, allow_missing=False,
                   force_init=True, allow_extra=False):
        self.init_params(None, arg_params, aux_params, allow_missing, force_init, allow_extra)

    def init_params(self, initializer=None, arg_params=None, aux_params=None,
                    allow_missing=False, force_init=False, allow_extra=False):
        if self.params_initialized and not force_init:
            return
        assert self.binded, "call bind before init_params"
        # Module.load stashed checkpoint params — consume them here so
        # load() -> bind() -> init_params()/fit() trains from the checkpoint,
        # not fresh init (reference Module.load, module.py:131-190).
        preloaded = getattr(self, "_preloaded", None)
        if preloaded is not None and arg_params is None and aux_params is None:
            arg_params, aux_params = preloaded
            initializer = None
            self._preloaded = None
        args, auxs = self._classified_named_tensors()
        if initializer is None and (arg_params is None and aux_params is None):
            initializer = init_mod.create("default")
        with torch.no_grad():
            if initializer is not None:
                self._apply_initializer(initializer)
            for name, tensor in list(args.items()) + list(auxs.items()):
                src = None
                if arg_params and name in arg_params:
                    src = arg_params[name]
                elif aux_params and name in aux_params:
                    src = aux_params[name]
                elif (arg_params or aux_params) and initializer is None and not allow_missing:
                    raise RuntimeError(f"parameter {name} missing from loaded params")
                if src is not None:
                    tensor.copy_(src.to(tensor.device, tensor.dtype))
        self.params_initialized = True

    def _apply_initializer(self, initializer):
        """Module-type-aware init (mxnet name-suffix dispatch equivalent):
        conv/linear weights -> initializer; biases -> 0; BN gamma/beta -> 1/0;
        BN stats -> 0/1."""
        for mname, m in self.symbol.named_modules():
            if isinstance(m, (torch.nn.Conv2d, torch.nn.Linear)) or m.__class__.__name__ in (
                "Conv2dNHWC", "LinearBF16"
            ):
                if getattr(m, "weight", None) is not None:
                    initializer(mname + ".weight", m.weight.data)
                if getattr(m, "bias", None) is not None:
                    m.bias.data.zero_()
            elif isinstance(m, torch.nn.modules.batchnorm._BatchNorm) or m.__class__.__name__ in (
                "BatchNorm2dNHWC",
            ):
                if getattr(m, "weight", None) is not None:
                    m.weight.data.fill_(1.0)
                if getattr(m, "bias", None) is not None:
                    m.bias.data.zero_()
                if getattr(m, "running_mean", None) is not None:
                    m.running_mean.zero_()
                if getattr(m, "running_var", None) is not None:
                    m.running_var.fill_(1.0)

    # ------------------------------------------------------------- optimizer
    def init_optimizer(self, kvstore="local", optimizer="sgd",
                       optimizer_params=(("learning_rate", 0.01),),
                       force_init=False, initialize_from_kvstore=False):
        """reference module.py:485-565. `initialize_from_kvstore=True` is the
        elastic joiner path: adopt cluster state instead of pushing init."""
        assert self.binded and self.params_initialized
        if self.optimizer_initialized and not force_init:
            return
        args, auxs = self._classified_named_tensors()
        if isinstance(kvstore, (str, type(None))):
            kv, update_on_kvstore = model_mod._create_kvstore(kvstore, 1, args)
        else:
            kv, update_on_kvstore = kvstore, True
        self._kvstore = kv
        self._update_on_kvstore = update_on_kvstore

        num_workers = kv.num_workers if kv is not None else 1
        if isinstance(optimizer, str):
            optimizer_params = dict(optimizer_params)
            optimizer_params.setdefault(
                "rescale_grad", 1.0 / (self._batch_size * num_workers)
            )
            if self._dtype in (torch.bfloat16, torch.float16):
                optimizer_params.setdefault("multi_precision", True)
            idx2name = {i: n for i, n in enumerate(args.keys())}
            optimizer = opt_create(optimizer, param_idx2name=idx2name, **optimizer_params)
        self._optimizer = optimizer
        self._updater = get_updater(optimizer)

        # register params with the kvstore; the store aliases the live
        # parameter tensors (device-resident, zero-copy — "device" comm).
        # A joiner (initialize_from_kvstore) skips the per-key init
        # broadcasts: survivors don't re-issue them after a re-form, so the
        # joiner adopts cluster state solely via _sync_full_state below —
        # both sides then run the identical collective sequence
        # (reference semantics: joiner pulls instead of pushing init,
        # kvstore_dist.h:205-224 / model.py:116-133).
        if kv is not None:
            if self._update_on_kvstore:
                kv.set_optimizer(optimizer)
            self._updater = kv._updater or self._updater
            if not (isinstance(kv, DistKVStore) and initialize_from_kvstore):
                for i, (name, p) in enumerate(args.items()):
                    kv.init(name, p.data)
                for name, b in auxs.items():
                    kv.init(name, b.data, exclude_update=True)
        # gradient bucketing (+ comm overlap when distributed). On GPU with
        # plain SGD+bf16, parameters are flattened so the whole optimizer
        # step is one fused HIP kernel per bucket.
        from ..optimizer.optimizer import SGD as _SGD

        on_gpu = self.device.type == "cuda"
        self._use_fused_sgd = (
            on_gpu
            and type(optimizer) is _SGD
            and self._dtype in (torch.bfloat16, torch.float16)
            and not self.fixed_param_names
        )
        if self._use_fused_sgd:
            from ..ops.hip import has_ext

            self._use_fused_sgd = has_ext()
        if not self._use_fused_sgd:
            # eagerly create optimizer state so elastic state broadcasts are
            # well-defined on every rank (momenta exist before any update)
            for i, (name, p) in enumerate(args.items()):
                st = self._updater.states.get(i)
                if st is None:
                    self._updater.states[i] = optimizer.create_state_multi_precision(i, p.data)
        if isinstance(kv, DistKVStore) or on_gpu:
            async_mode = isinstance(kv, DistKVStore) and "async" in kv.type
            self._bucketer = GradBucketer(list(args.values()),
                                          flatten_params=self._use_fused_sgd,
                                          async_mode=async_mode)
            comp = getattr(kv, "_compression", None)
            if isinstance(kv, DistKVStore) and comp is not None:
                self._bucketer.set_compression(comp.threshold)
            if isinstance(kv, DistKVStore) and initialize_from_kvstore:
                self._sync_full_state()
        self.optimizer_initialized = True

    def _sync_full_state(self):
        """Broadcast weights+aux+optimizer state from rank 0 (joiner adoption;
        reference model.py:116-133 interleaved param+aux pull)."""
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return
        args, auxs = self._classified_named_tensors()
        if self._use_fused_sgd and self._bucketer is not None:
            # params are views of the flat buffers: one broadcast per bucket
            tensors: List[torch.Tensor] = list(self._bucketer.state_tensors())
            tensors += [b.data for b in auxs.values()]
        else:
            tensors = [p.data for p in args.values()]
            tensors += [b.data for b in auxs.values()]
            for i in sorted(self._updater.states.keys()):
                tensors += _flatten_state(self._updater.states[i])
        with torch.no_grad():
            for t in tensors:
                dist.broadcast(t, src=0)

    # ------------------------------------------------------- train internals
    def _to_device(self, t: torch.Tensor, is_data: bool) -> torch.Tensor:
        t = t.to(self.device, non_blocking=True)
        if is_data and t.dtype.is_floating_point:
            t = t.to(self._dtype)
            if self._channels_last and t.dim() == 4:
                t = t.contiguous(memory_format=torch.channels_last)
        return t

    def forward(self, data_batch: DataBatch, is_train: Optional[bool] = None):
        assert self.binded and self.params_initialized
        if is_train is None:
            is_train = self.for_training
        self.symbol.train(is_train)
        data = [self._to_device(d, True) for d in data_batch.data]
        self._labels = [self._to_device(l, False) for l in (data_batch.label or [])]
        with torch.enable_grad() if is_train else torch.no_grad():
            out = self.symbol(*data)
        self._outputs = list(_as_list(out))
        if is_train and self._labels:
            # SoftmaxOutput semantics: per-sample CE, grad = (p - onehot);
            # sum-reduction so grads are batch sums; the optimizer's
            # rescale_grad = 1/(B*W) turns the summed all-reduce into the
            # global-batch mean (reference optimizer.py:374-400).
            logits = self._outputs[0]
            label = self._labels[0].reshape(-1).long()
            self._loss = torch.nn.functional.cross_entropy(
                logits.float(), label, reduction="sum"
            )
        else:
            self._loss = None

    def backward(self, out_grads=None):
        assert self.for_training
        if self._bucketer is not None:
            self._bucketer.zero_grad()
        else:
            self.symbol.zero_grad(set_to_none=False)
        if self._loss is not None:
            self._loss.backward()
        elif out_grads is not None:
            torch.autograd.backward(self._outputs, [g.to(self.device) for g in _as_list(out_grads)])

    def forward_backward(self, data_batch: DataBatch):
        self.forward(data_batch, is_train=True)
        self.backward()

    def update(self, update_aux_params: bool = False):
        """Apply the optimizer after gradient reduction (reference
        module.py:685-720 / model.py:183)."""
        assert self.binded and self.params_initialized and self.optimizer_initialized
        args, auxs = self._classified_named_tensors()
        if self._bucketer is not None:
            self._bucketer.finish()  # join async all-reduces (sum over workers)
            if self._use_fused_sgd:
                opt = self._optimizer
                opt.num_update += 1
                lr = opt.lr_scheduler(opt.num_update) if opt.lr_scheduler else opt.lr
                hyper = None
                if self._capture_mode:
                    hyper = self._update_hyper(lr)
                self._bucketer.fused_sgd_step(
                    lr, getattr(opt, "momentum", 0.0), opt.wd, opt.rescale_grad,
                    opt.clip_gradient or 0.0, hyper=hyper,
                )
            else:
                grads = (
                    self._bucketer.reduced_views()
                    if self._bucketer.async_mode
                    else [p.grad for p in args.values()]
                )
                for i, (name, p) in enumerate(args.items()):
                    if name in self.fixed_param_names or grads[i] is None:
                        continue
                    self._updater(i, grads[i], p.data)
        elif isinstance(self._kvstore, LocalKVStore) or self._kvstore is None:
            for i, (name, p) in enumerate(args.items()):
                if name in self.fixed_param_names or p.grad is None:
                    continue
                self._updater(i, p.grad, p.data)
        else:
            for i, (name, p) in enumerate(args.items()):
                if name in self.fixed_param_names or p.grad is None:
                    continue
                self._kvstore.push(name, p.grad)
                self._kvstore.pull(name, out=p.data)
        if update_aux_params:
            self.store_aux_params()

    # -------------------------------------------------- hipGraph step replay
    def _update_hyper(self, lr: float) -> torch.Tensor:
        opt = self._optimizer
        if self._hyper_dev is None:
            self._hyper_dev = torch.zeros(5, dtype=torch.float32, device=self.device)
            self._hyper_host = torch.zeros(5, dtype=torch.float32, pin_memory=True)
        self._hyper_host[0] = lr
        self._hyper_host[1] = getattr(opt, "momentum", 0.0)
        self._hyper_host[2] = opt.wd
        self._hyper_host[3] = opt.rescale_grad
        self._hyper_host[4] = opt.clip_gradient or 0.0
        self._hyper_dev.copy_(self._hyper_host, non_blocking=True)
        return self._hyper_dev

    def graphed_step(self, data_batch: DataBatch):
        """One fwd+bwd+update captured in a hipGraph and replayed (launch-
        bound inner loops belong in hipGraphs — the MI355X-native replacement
        for the reference's executor op-segment bulking). Requires static
        shapes; falls back to eager when capture is unavailable.

        Multi-rank: the bucketer's RCCL all-reduces are captured into the
        graph (ProcessGroupNCCL supports hipGraph capture)."""
        if not (self.device.type == "cuda" and self._use_fused_sgd):
            self.forward_backward(data_batch)
            self.update()
            return
        if self._graph is None:
            self._capture_mode = True
            data = [self._to_device(d, True) for d in data_batch.data]
            labels = [self._to_device(l, False) for l in (data_batch.label or [])]
            self._static_inputs = (data, labels)
            static_batch = DataBatch(data=data, label=labels)
            # capture the LAYERWISE block path: replay of it is validated
            # (tools/graph_numerics.py); the fused-block manual backward is
            # not yet capture-safe and diverges under replay.
            import os
            prev_fb = os.environ.get("DTMX_FUSED_BLOCK")
            os.environ["DTMX_FUSED_BLOCK"] = "0"
            try:
                # warmup on a side stream (allocator + autograd graph priming)
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(3):
                        self.forward_backward(static_batch)
                        self.update()
                torch.cuda.current_stream().wait_stream(side)
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self.forward_backward(static_batch)
                    self.update()
            finally:
                if prev_fb is None:
                    os.environ.pop("DTMX_FUSED_BLOCK", None)
                else:
                    os.environ["DTMX_FUSED_BLOCK"] = prev_fb
            self._graph = g
            return
        data, labels = self._static_inputs
        with torch.no_grad():
            for dst, src in zip(data, data_batch.data):
                if dst.data_ptr() != src.data_ptr():
                    dst.copy_(self._to_device(src, True), non_blocking=True)
            for dst, src in zip(labels, data_batch.label or []):
                if dst.data_ptr() != src.data_ptr():
                    dst.copy_(self._to_device(src, False), non_blocking=True)
        # increment num_update BEFORE deriving lr, matching the eager
        # update() path — otherwise replayed steps run one lr-update stale.
        self._optimizer.num_update += 1
        self._update_hyper(
            self._optimizer.lr_scheduler(self._optimizer.num_update)
            if self._optimizer.lr_scheduler else self._optimizer.lr
        )
        self._graph.replay()

    def store_aux_params(self):
        """Cluster-average BN running stats (reference module.py:668-683 ->
        aux-key averaging on the server)."""
        _, auxs = self._classified_named_tensors()
        if isinstance(self._kvstore, DistKVStore) and self._kvstore.num_workers > 1:
            with torch.no_grad():
                for name, b in auxs.items():
                    t = b.data.float()
                    dist.all_reduce(t, op=dist.ReduceOp.SUM)
                    b.data.copy_((t / self._kvstore.num_workers).to(b.dtype))
        elif self._kvstore is not None and not isinstance(self._kvstore, DistKVStore):
            for name, b in auxs.items():
                self._kvstore.push(name, b.data)

    def update_metric(self, eval_metric, labels, pre_sliced=False):
        if not self._outputs:
            return
        eval_metric.update(labels, [o.detach() for o in self._outputs])

    def get_outputs(self):
        return self._outputs

    # ------------------------------------------------------------------- fit
    def get_iterator(self, kv) -> Optional[DataIter]:
        """Re-create the data iterator for the current (rank, num_workers)
        (reference module.py:198-202 / ETDataIterator fit.py:31-44)."""
        if self._data_iterator is None:
            return None
        return self._data_iterator(kv)

    def fit(self, train_data, eval_data=None, eval_metric="acc",
            epoch_end_callback=None, batch_end_callback=None, kvstore="local",
            optimizer="sgd", optimizer_params=(("learning_rate", 0.01),),
            eval_end_callback=None, initializer=None, arg_params=None,
            aux_params=None, allow_missing=False, force_rebind=False,
            force_init=False, begin_epoch=0, num_epoch=None,
            validation_metric=None, data_iterator=None, elastic_training=None):
        assert num_epoch is not None
        is_new_worker = os.environ.get("NEW_WORKER", "0") == "1"
        if is_new_worker:
            begin_epoch = int(os.environ.get("EPOCH_BEGIN", begin_epoch))
        self._data_iterator = data_iterator

        if callable(train_data) and not isinstance(train_data, DataIter):
            self._data_iterator = train_data
            train_data = None

        if not self.binded:
            raise RuntimeError("bind() before fit() (dtmx requires explicit bind)")
        self.init_params(initializer, arg_params, aux_params, allow_missing,
                         force_init=force_init)
        self.init_optimizer(kvstore=kvstore, optimizer=optimizer,
                            optimizer_params=optimizer_params, force_init=force_init,
                            initialize_from_kvstore=is_new_worker)
        kv = self._kvstore
        if elastic_training is None:
            elastic_training = os.environ.get("ELASTIC_TRAINING_ENABLED", "0").lower() in ("1", "true")

        if train_data is None:
            train_data = self.get_iterator(kv)
        assert train_data is not None, "no train_data and no data_iterator factory"

        # A joiner's EPOCH_BEGIN env is a stale hint (written at an earlier
        # barrier). Its rendezvous completed at the survivors' barrier, which
        # wrote the authoritative cluster epoch — adopt it so the local epoch
        # counter aligns and every worker terminates at the same epoch
        # (otherwise the joiner trails and strands at a barrier after the
        # survivors exit).
        if (is_new_worker and isinstance(kv, DistKVStore)
                and getattr(kv, "_elastic", None) is not None
                and getattr(kv._elastic, "store", None) is not None):
            try:
                begin_epoch = int(kv._elastic.store.get("cluster/epoch"))
            except Exception:
                pass

        eval_metric = metric_mod.create(eval_metric)
        if validation_metric is None:
            validation_metric = eval_metric

        for epoch in range(begin_epoch, num_epoch):
            # -- membership-change barrier at epoch top (base_module.py:540) --
            if elastic_training and isinstance(kv, DistKVStore):
                if is_new_worker and epoch == begin_epoch:
                    pass  # joiner completed the roster at startup
                else:
                    old_w = kv.num_workers
                    changed = kv._membership_change_barrier({"EPOCH_BEGIN": str(epoch)})
                    if changed:
                        new_w = kv.num_workers
                        self.logger.info(
                            "Node[%d] membership changed: %d -> %d workers",
                            kv.rank, old_w, new_w,
                        )
                        # dynamic-minibatch LR rescale (the paper's recipe)
                        self._optimizer.rescale_grad = 1.0 / (self._batch_size * new_w)
                        sched = getattr(self._optimizer, "lr_scheduler", None)
                        if isinstance(sched, WarmupScheduler) and new_w > old_w:
                            sched.rearm(self._optimizer.num_update)
                        if self._bucketer is not None:
                            self._bucketer.rebuild_after_membership_change()
                        self._sync_full_state()
                        new_iter = self.get_iterator(kv)
                        if new_iter is not None:
                            train_data = new_iter

            tic = time.time()
            eval_metric.reset()
            train_data.reset()
            nbatch = 0
            for data_batch in train_data:
                self.forward_backward(data_batch)
                self.update()
                self.update_metric(eval_metric, self._labels)
                if batch_end_callback is not None:
                    p = BatchEndParam(epoch=epoch, nbatch=nbatch, eval_metric=eval_metric,
                                      locals=locals())
                    for cb in _as_list(batch_end_callback):
                        cb(p)
                nbatch += 1

            for name, val in eval_metric.get_name_value():
                self.logger.info("Epoch[%d] Train-%s=%f", epoch, name, val)
            self.logger.info("Epoch[%d] Time cost=%.3f", epoch, time.time() - tic)

            # -- epoch-end aux sync (base_module.py:603-605) --
            self.store_aux_params()

            if epoch_end_callback is not None:
                arg_p, aux_p = self.get_params()
                for cb in _as_list(epoch_end_callback):
                    cb(epoch, self.symbol, arg_p, aux_p)

            if eval_data is not None:
                res = self.score(eval_data, validation_metric)
                for name, val in res:
                    self.logger.info("Epoch[%d] Validation-%s=%f", epoch, name, val)
            is_new_worker = False  # only the first epoch is special for a joiner

    # ------------------------------------------------------------ evaluation
    def score(self, eval_data: DataIter, eval_metric, num_batch=None):
        eval_metric = metric_mod.create(eval_metric)
        eval_metric.reset()
        eval_data.reset()
        for nbatch, batch in enumerate(eval_data):
            if num_batch is not None and nbatch == num_batch:
                break
            self.forward(batch, is_train=False)
            self.update_metric(eval_metric, [self._to_device(l, False) for l in batch.label])
        return eval_metric.get_name_value()

    def predict(self, eval_data: DataIter, num_batch=None):
        outs = []
        eval_data.reset()
        for nbatch, batch in enumerate(eval_data):
            if num_batch is not None and nbatch == num_batch:
                break
            self.forward(batch, is_train=False)
            outs.append(self._outputs[0].detach().cpu())
        return torch.cat(outs, dim=0)

    # ----------------------------------------------------------- checkpoints
    def save_optimizer_states(self, fname: str):
        """reference module.py:839-868 save_optimizer_states."""
        import pickle

        if self._use_fused_sgd and self._bucketer is not None:
            payload = {
                "fused": True,
                "master": [t.cpu() for t in self._bucketer.flat_master],
                "mom": [t.cpu() for t in self._bucketer.flat_mom],
                "num_update": self._optimizer.num_update,
            }
            with open(fname, "wb") as f:
                pickle.dump(payload, f)
        elif self._kvstore is not None:
            self._kvstore._updater = self._updater
            self._kvstore._optimizer = self._optimizer
            self._kvstore.save_optimizer_states(fname)

    def load_optimizer_states(self, fname: str):
        import pickle

        with open(fname, "rb") as f:
            payload = pickle.load(f)
        if isinstance(payload, dict) and payload.get("fused"):
            assert self._use_fused_sgd and self._bucketer is not None
            with torch.no_grad():
                for dst, src in zip(self._bucketer.flat_master, payload["master"]):
                    dst.copy_(src.to(dst.device))
                for dst, src in zip(self._bucketer.flat_mom, payload["mom"]):
                    dst.copy_(src.to(dst.device))
                for wb, mb in zip(self._bucketer.flat_w, self._bucketer.flat_master):
                    wb.copy_(mb.to(wb.dtype))
            self._optimizer.num_update = payload.get("num_update", 0)
        elif self._kvstore is not None:
            self._kvstore._updater = self._updater
            self._kvstore.load_optimizer_states(fname)

    def save_checkpoint(self, prefix: str, epoch: int, save_optimizer_states=False):
        arg_p, aux_p = self.get_params()
        model_mod.save_checkpoint(prefix, epoch, self.symbol, arg_p, aux_p)
        if save_optimizer_states:
            self.save_optimizer_states("%s-%04d.states" % (prefix, epoch))

    @staticmethod
    def load(prefix: str, epoch: int, symbol_builder, **kwargs) -> "Module":
        _, arg_params, aux_params = model_mod.load_checkpoint(prefix, epoch)
        mod = Module(symbol_builder, **kwargs)
        mod._preloaded = (arg_params, aux_params)
        return mod


def _flatten_state(state) -> List[torch.Tensor]:
    if state is None:
        return []
    if isinstance(state, torch.Tensor):
        return [state]
    out: List[torch.Tensor] = []
    for s in state:
        out += _flatten_state(s)
    return out
