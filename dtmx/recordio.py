"""RecordIO python API (reference python/mxnet/recordio.py): IRHeader
pack/unpack and the MXRecordIO / MXIndexedRecordIO file classes, riding on
the native reader/writer (dtmx/csrc/recordio.cpp — dmlc magic-framed
records, reference dmlc-core recordio.h)."""
from __future__ import annotations

import struct
from collections import namedtuple

IRHeader = namedtuple("IRHeader", ["flag", "label", "id", "id2"])
_IR_FORMAT = "<IfQQ"
_IR_SIZE = struct.calcsize(_IR_FORMAT)


def pack(header: IRHeader, s: bytes) -> bytes:
    """Pack a header + payload into a record blob (reference recordio.py:pack).
    An array label is stored ahead of the payload with flag = len(label)."""
    label = header.label
    if hasattr(label, "__len__"):
        import numpy as np
        arr = np.asarray(label, dtype=np.float32)
        header = IRHeader(len(arr), 0.0, header.id, header.id2)
        s = arr.tobytes() + s
    return struct.pack(_IR_FORMAT, *header) + s


def unpack(s: bytes):
    """Unpack a record blob into (IRHeader, payload)."""
    header = IRHeader(*struct.unpack(_IR_FORMAT, s[:_IR_SIZE]))
    s = s[_IR_SIZE:]
    if header.flag > 0:
        import numpy as np
        n = header.flag
        label = np.frombuffer(s[: n * 4], dtype=np.float32).copy()
        header = header._replace(label=label)
        s = s[n * 4:]
    return header, s


class MXRecordIO:
    """Sequential RecordIO file. 'r' reads via the native reader; 'w' buffers
    records and writes the magic-framed file on close()."""

    def __init__(self, uri: str, flag: str):
        assert flag in ("r", "w"), flag
        self.uri = uri
        self.flag = flag
        self._cursor = 0
        self._pending = []
        self._reader = None
        self.open()

    def open(self):
        if self.flag == "r":
            from .ops.hip import get_ext
            self._reader = get_ext().RecordIOReader(self.uri)
        self._cursor = 0

    def close(self):
        if self.flag == "w" and self._pending is not None:
            from .ops.hip import get_ext
            get_ext().write_recordio(self.uri, self._pending)
            self._pending = None
        self._reader = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def reset(self):
        self._cursor = 0

    def write(self, buf: bytes):
        assert self.flag == "w", "file opened for reading"
        self._pending.append(bytes(buf))

    def read(self):
        assert self.flag == "r", "file opened for writing"
        if self._cursor >= len(self._reader):
            return None
        rec = self._reader.read(self._cursor)
        self._cursor += 1
        return bytes(rec)

    def __len__(self):
        if self._reader is not None:
            return len(self._reader)
        return len(self._pending or [])


class MXIndexedRecordIO(MXRecordIO):
    """RecordIO with a sidecar .idx of `key<TAB>position` lines
    (reference recordio.py:MXIndexedRecordIO); positions here are record
    ordinals — the native reader is random-access by index."""

    def __init__(self, idx_path: str, uri: str, flag: str):
        self.idx_path = idx_path
        self.idx = {}
        super().__init__(uri, flag)
        if flag == "r":
            with open(idx_path) as f:
                for line in f:
                    parts = line.split("\t")
                    if len(parts) >= 2:
                        self.idx[int(parts[0])] = int(parts[1])

    def close(self):
        if self.flag == "w" and self._pending is not None:
            with open(self.idx_path, "w") as f:
                for k, pos in self.idx.items():
                    f.write(f"{k}\t{pos}\n")
        super().close()

    def keys(self):
        return list(self.idx)

    def read_idx(self, key):
        assert self.flag == "r"
        return bytes(self._reader.read(self.idx[key]))

    def write_idx(self, key, buf: bytes):
        self.idx[key] = len(self._pending)
        self.write(buf)
