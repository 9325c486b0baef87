"""Data iterators (reference python/mxnet/io/, src/io/).

The reference's iterator contract consumed by Module.fit:
  - `provide_data` / `provide_label`: [(name, shape)] with batch dim included
  - `reset()`, `iter_next()`, `next() -> DataBatch(data=[...], label=[...], pad=n)`
  - distributed sharding via (part_index, num_parts) — re-derived after an
    elastic membership change (reference ETDataIterator, common/fit.py:31-44).
"""
from __future__ import annotations

import os
import struct
from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch


class DataDesc(tuple):
    def __new__(cls, name, shape, dtype=torch.float32):
        ret = super().__new__(cls, (name, shape))
        ret.name = name
        ret.shape = shape
        ret.dtype = dtype
        return ret


class DataBatch:
    def __init__(self, data: List[torch.Tensor], label: Optional[List[torch.Tensor]] = None,
                 pad: int = 0, index=None, provide_data=None, provide_label=None):
        self.data = data
        self.label = label
        self.pad = pad
        self.index = index
        self.provide_data = provide_data
        self.provide_label = provide_label


class DataIter:
    def __init__(self, batch_size: int = 0):
        self.batch_size = batch_size

    def __iter__(self):
        return self

    def reset(self):
        pass

    def next(self) -> DataBatch:
        raise NotImplementedError

    def __next__(self):
        return self.next()

    @property
    def provide_data(self):
        raise NotImplementedError

    @property
    def provide_label(self):
        raise NotImplementedError


class NDArrayIter(DataIter):
    """In-memory iterator (reference python/mxnet/io/io.py NDArrayIter) with
    (part_index, num_parts) sharding for distributed training."""

    def __init__(self, data, label=None, batch_size=1, shuffle=False,
                 last_batch_handle="pad", data_name="data", label_name="softmax_label",
                 part_index: int = 0, num_parts: int = 1):
        super().__init__(batch_size)
        self.data = self._init_data(data, data_name)
        self.label = self._init_data(label, label_name) if label is not None else []
        self.shuffle = shuffle
        self.last_batch_handle = last_batch_handle
        n_total = self.data[0][1].shape[0]
        # shard rows [part_index*per, ...): equal slices, like reference io.py num_parts
        per = n_total // num_parts
        start = part_index * per
        end = start + per if part_index < num_parts - 1 else n_total
        self._sel_base = np.arange(start, end)
        self.num_data = len(self._sel_base)
        if self.num_data < batch_size:
            raise ValueError("batch_size larger than shard size")
        self.cursor = -batch_size
        self._order = self._sel_base.copy()
        self.reset()

    @staticmethod
    def _init_data(data, default_name) -> List[Tuple[str, torch.Tensor]]:
        if data is None:
            return []
        if isinstance(data, (np.ndarray, torch.Tensor)):
            data = {default_name: data}
        elif isinstance(data, (list, tuple)):
            data = {f"{default_name}{i if i else ''}": d for i, d in enumerate(data)}
        out = []
        for k, v in data.items():
            t = torch.as_tensor(np.asarray(v)) if not isinstance(v, torch.Tensor) else v
            out.append((k, t))
        return out

    @property
    def provide_data(self):
        return [DataDesc(k, (self.batch_size,) + tuple(v.shape[1:]), v.dtype)
                for k, v in self.data]

    @property
    def provide_label(self):
        return [DataDesc(k, (self.batch_size,) + tuple(v.shape[1:]), v.dtype)
                for k, v in self.label]

    def reset(self):
        if self.shuffle:
            self._order = self._sel_base[np.random.permutation(self.num_data)]
        self.cursor = -self.batch_size

    def iter_next(self) -> bool:
        self.cursor += self.batch_size
        return self.cursor < self.num_data

    def next(self) -> DataBatch:
        if not self.iter_next():
            raise StopIteration
        idx = self._order[self.cursor : self.cursor + self.batch_size]
        pad = 0
        if len(idx) < self.batch_size:
            if self.last_batch_handle == "discard":
                raise StopIteration
            pad = self.batch_size - len(idx)
            idx = np.concatenate([idx, self._order[:pad]])
        ti = torch.from_numpy(idx.astype(np.int64))
        return DataBatch(
            data=[v[ti] for _, v in self.data],
            label=[v[ti] for _, v in self.label],
            pad=pad,
            provide_data=self.provide_data,
            provide_label=self.provide_label,
        )


class SyntheticDataIter(DataIter):
    """Benchmark iterator (reference example/image-classification/common/
    data.py:98-131): fixed random batch replayed; no host->device copies in
    steady state when `ctx` is a GPU."""

    def __init__(self, num_classes, data_shape, max_iter, dtype=torch.float32,
                 label_name="softmax_label", device=None, layout: str = "NCHW"):
        super().__init__(data_shape[0])
        self.cur_iter = 0
        self.max_iter = max_iter
        self.dtype = dtype
        self.data_shape = tuple(data_shape)
        label = np.random.randint(0, num_classes, (self.batch_size,))
        data = np.random.uniform(-1, 1, data_shape).astype(np.float32)
        self._data = torch.from_numpy(data).to(dtype)
        self._label = torch.from_numpy(label.astype(np.float32))
        if layout == "NHWC" and len(self.data_shape) == 4:
            self._data = self._data.contiguous(memory_format=torch.channels_last)
        if device is not None:
            self._data = self._data.to(device)
            self._label = self._label.to(device)
        self.label_name = label_name

    @property
    def provide_data(self):
        return [DataDesc("data", self.data_shape, self.dtype)]

    @property
    def provide_label(self):
        return [DataDesc(self.label_name, (self.batch_size,), torch.float32)]

    def next(self):
        self.cur_iter += 1
        if self.cur_iter > self.max_iter:
            raise StopIteration
        return DataBatch(
            data=[self._data], label=[self._label], pad=0,
            provide_data=self.provide_data, provide_label=self.provide_label,
        )

    def reset(self):
        self.cur_iter = 0


class MNISTIter(DataIter):
    """IDX-format MNIST reader (reference src/io/iter_mnist.cc). Shards by
    (part_index, num_parts). Falls back to deterministic synthetic digits when
    the idx files are absent (this environment has no network)."""

    def __init__(self, image: str = "", label: str = "", batch_size: int = 64,
                 shuffle: bool = True, flat: bool = False, part_index: int = 0,
                 num_parts: int = 1, seed: int = 0, num_examples: int = 2048):
        super().__init__(batch_size)
        if image and os.path.exists(image):
            img = self._read_idx(image).astype(np.float32) / 255.0
            lab = self._read_idx(label).astype(np.float32)
        else:
            rng = np.random.RandomState(seed)
            lab = rng.randint(0, 10, (num_examples,)).astype(np.float32)
            protos = np.random.RandomState(1234).randn(10, 28, 28).astype(np.float32)
            img = (protos[lab.astype(np.int64)]
                   + 0.5 * rng.randn(num_examples, 28, 28).astype(np.float32))
        if flat:
            img = img.reshape(len(img), -1)
        else:
            img = img.reshape(len(img), 1, 28, 28)
        self._iter = NDArrayIter(
            {"data": img}, {"softmax_label": lab}, batch_size, shuffle=shuffle,
            part_index=part_index, num_parts=num_parts,
        )

    @staticmethod
    def _read_idx(path: str) -> np.ndarray:
        with open(path, "rb") as f:
            zero, dt, ndim = struct.unpack(">HBB", f.read(4))
            shape = struct.unpack(f">{ndim}I", f.read(4 * ndim))
            return np.frombuffer(f.read(), dtype=np.uint8).reshape(shape)

    @property
    def provide_data(self):
        return self._iter.provide_data

    @property
    def provide_label(self):
        return self._iter.provide_label

    def reset(self):
        self._iter.reset()

    def next(self):
        return self._iter.next()


class CSVIter(DataIter):
    """CSV reader (reference src/io/iter_csv.cc)."""

    def __init__(self, data_csv: str, data_shape, label_csv: Optional[str] = None,
                 label_shape=(1,), batch_size: int = 1, part_index: int = 0,
                 num_parts: int = 1):
        super().__init__(batch_size)
        data = np.loadtxt(data_csv, delimiter=",", dtype=np.float32, ndmin=2)
        data = data.reshape((-1,) + tuple(data_shape))
        label = None
        if label_csv:
            label = np.loadtxt(label_csv, delimiter=",", dtype=np.float32, ndmin=2)
            label = label.reshape((-1,) + tuple(label_shape)).squeeze(-1)
        self._iter = NDArrayIter({"data": data},
                                 {"softmax_label": label} if label is not None else None,
                                 batch_size, part_index=part_index, num_parts=num_parts)

    @property
    def provide_data(self):
        return self._iter.provide_data

    @property
    def provide_label(self):
        return self._iter.provide_label

    def reset(self):
        self._iter.reset()

    def next(self):
        return self._iter.next()


class ResizeIter(DataIter):
    """Caps an iterator at `size` batches per epoch (reference io.py ResizeIter;
    used by fit.py epoch_size = num_examples/num_workers/batch — the quantity
    recomputed after membership changes)."""

    def __init__(self, data_iter: DataIter, size: int, reset_internal: bool = True):
        super().__init__(data_iter.batch_size)
        self.data_iter = data_iter
        self.size = size
        self.reset_internal = reset_internal
        self.cur = 0

    @property
    def provide_data(self):
        return self.data_iter.provide_data

    @property
    def provide_label(self):
        return self.data_iter.provide_label

    def reset(self):
        self.cur = 0
        if self.reset_internal:
            self.data_iter.reset()

    def next(self):
        if self.cur == self.size:
            raise StopIteration
        try:
            batch = self.data_iter.next()
        except StopIteration:
            self.data_iter.reset()
            batch = self.data_iter.next()
        self.cur += 1
        return batch


class PrefetchingIter(DataIter):
    """Background-thread prefetcher (reference src/io/iter_prefetcher.h)."""

    def __init__(self, data_iter: DataIter, capacity: int = 2):
        super().__init__(data_iter.batch_size)
        import queue
        import threading

        self.data_iter = data_iter
        self._queue: "queue.Queue" = queue.Queue(maxsize=capacity)
        self._stop = threading.Event()
        self._wake = threading.Event()
        # generation counter: reset() bumps it; the worker resets the inner
        # iterator when it observes the bump, and every queued batch carries
        # the generation it was produced under so the consumer can discard
        # pre-reset leftovers. This makes reset() correct even when the
        # consumer abandons an epoch mid-way (score(num_batch=...),
        # ResizeIter caps) — the old design only handled reset-at-
        # StopIteration and could deliver stale batches otherwise.
        self._gen = 0
        self._thread = threading.Thread(target=self._worker, daemon=True)
        self._thread.start()

    def _worker(self):
        import queue

        worker_gen = 0
        while not self._stop.is_set():
            if worker_gen != self._gen:
                worker_gen = self._gen
                self.data_iter.reset()
                continue
            try:
                batch = self.data_iter.next()
            except StopIteration:
                batch = None
            # put, abandoning the item if a reset or stop arrives while full
            while not self._stop.is_set() and worker_gen == self._gen:
                try:
                    self._queue.put((worker_gen, batch), timeout=0.05)
                    break
                except queue.Full:
                    continue
            if batch is None and worker_gen == self._gen:
                # parked at epoch end until reset() or close()
                self._wake.wait()
                self._wake.clear()

    @property
    def provide_data(self):
        return self.data_iter.provide_data

    @property
    def provide_label(self):
        return self.data_iter.provide_label

    def reset(self):
        import queue

        self._gen += 1
        self._wake.set()
        # drain stale entries so a blocked worker can make progress; any
        # stale item that races in after this drain is filtered in next()
        while True:
            try:
                self._queue.get_nowait()
            except queue.Empty:
                break

    def next(self):
        while True:
            gen, batch = self._queue.get()
            if gen != self._gen:
                continue  # produced before the last reset()
            if batch is None:
                raise StopIteration
            return batch

    def close(self):
        self._stop.set()
        self._wake.set()


class ImageRecordIter(DataIter):
    """RecordIO-backed image iterator on the native C++ pipeline
    (reference src/io/iter_image_recordio_2.cc; dtmx/csrc/recordio.cpp):
    threaded record parsing + JPEG decode (libjpeg) + resize/random-crop/
    mirror augmentation + batch assembly + bounded prefetch queue, sharded
    by (part_index, num_parts). Records carry JPEG payloads or raw uint8
    HWC of the target shape (tools/im2rec.py packs both)."""

    def __init__(self, path_imgrec: str, data_shape, batch_size: int,
                 shuffle: bool = False, part_index: int = 0, num_parts: int = 1,
                 preprocess_threads: int = 4, prefetch_buffer: int = 4,
                 seed: int = 0, label_name: str = "softmax_label",
                 rand_crop: bool = False, rand_mirror: bool = False,
                 resize: int = 0, mean_r: float = 0.0, mean_g: float = 0.0,
                 mean_b: float = 0.0, std_r: float = 1.0, std_g: float = 1.0,
                 std_b: float = 1.0, **kwargs):
        super().__init__(batch_size)
        from dtmx.ops.hip import require_ext

        ext = require_ext()
        self._reader = ext.RecordIOReader(path_imgrec)
        # data_shape is CHW (reference convention); records store HWC raw
        c, h, w = data_shape
        self._chw = (c, h, w)
        self._loader = ext.RecordBatchLoader(
            self._reader, batch_size, [h, w, c], part_index, num_parts,
            shuffle, preprocess_threads, prefetch_buffer, seed,
            rand_crop, rand_mirror, resize,
        )
        self.label_name = label_name
        # channel-wise (x - mean) / std (reference ImageRecordIter
        # mean_r/g/b + std_r/g/b params, image_aug_default.cc)
        if (mean_r, mean_g, mean_b) != (0.0, 0.0, 0.0) or \
                (std_r, std_g, std_b) != (1.0, 1.0, 1.0):
            self._mean = torch.tensor([mean_r, mean_g, mean_b][:c]).reshape(1, c, 1, 1)
            self._std = torch.tensor([std_r, std_g, std_b][:c]).reshape(1, c, 1, 1)
        else:
            self._mean = self._std = None

    @property
    def provide_data(self):
        return [DataDesc("data", (self.batch_size,) + self._chw, torch.float32)]

    @property
    def provide_label(self):
        return [DataDesc(self.label_name, (self.batch_size,), torch.float32)]

    def reset(self):
        self._loader.reset()

    def next(self):
        out = self._loader.next()
        if not out:
            raise StopIteration
        data, label = out
        data = data.permute(0, 3, 1, 2)  # HWC records -> logical NCHW (NHWC memory)
        if self._mean is not None:
            data = (data - self._mean) / self._std
        return DataBatch(data=[data], label=[label], pad=0,
                         provide_data=self.provide_data,
                         provide_label=self.provide_label)
