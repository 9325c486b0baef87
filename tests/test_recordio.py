"""Native RecordIO pipeline (reference dmlc RecordIO + iter_image_recordio_2):
format round trip, threaded loader, sharding."""
import struct

import numpy as np
import pytest
import torch

from dtmx.ops.hip import get_ext

pytestmark = pytest.mark.skipif(get_ext() is None, reason="ext not built")


def _write(tmp_path, n=64, h=8, w=8, c=3):
    ext = get_ext()
    rng = np.random.RandomState(0)
    x = rng.randint(0, 256, (n, h, w, c), dtype=np.uint8)
    y = rng.randint(0, 10, (n,))
    recs = [struct.pack("<IfQQ", 0, float(y[i]), i, 0) + x[i].tobytes()
            for i in range(n)]
    path = str(tmp_path / "d.rec")
    ext.write_recordio(path, recs)
    return path, x, y


def test_reader_roundtrip(tmp_path):
    ext = get_ext()
    path, x, y = _write(tmp_path)
    r = ext.RecordIOReader(path)
    assert len(r) == 64
    rec = r.read(3)
    flag, label, rid, _ = struct.unpack("<IfQQ", rec[:24])
    assert label == float(y[3]) and rid == 3
    assert rec[24:] == x[3].tobytes()


def test_image_record_iter(tmp_path):
    from dtmx.io import ImageRecordIter

    path, x, y = _write(tmp_path)
    it = ImageRecordIter(path, (3, 8, 8), batch_size=16, shuffle=False,
                         preprocess_threads=2)
    batches = list(it)
    assert len(batches) == 4
    b0 = batches[0]
    assert b0.data[0].shape == (16, 3, 8, 8)
    # first record round-trips through the loader (label + scaled pixels)
    assert b0.label[0][0].item() == float(y[0])
    got = (b0.data[0][0].permute(1, 2, 0) * 255).round().to(torch.uint8)
    assert torch.equal(got, torch.from_numpy(x[0]))
    it.reset()
    assert len(list(it)) == 4


def test_sharding(tmp_path):
    from dtmx.io import ImageRecordIter

    path, x, y = _write(tmp_path)
    it0 = ImageRecordIter(path, (3, 8, 8), 8, part_index=0, num_parts=2)
    it1 = ImageRecordIter(path, (3, 8, 8), 8, part_index=1, num_parts=2)
    l0 = [b.label[0][0].item() for b in it0]
    l1 = [b.label[0][0].item() for b in it1]
    assert len(l0) == 4 and len(l1) == 4
    assert l0[0] == float(y[0]) and l1[0] == float(y[32])


def test_recordio_python_api(tmp_path):
    """mx.recordio surface: pack/unpack + MXRecordIO + MXIndexedRecordIO
    (reference python/mxnet/recordio.py)."""
    import numpy as np
    from dtmx import recordio

    # scalar and array labels round-trip through pack/unpack
    h = recordio.IRHeader(0, 7.0, 3, 0)
    hdr, payload = recordio.unpack(recordio.pack(h, b"abc"))
    assert hdr.label == 7.0 and payload == b"abc"
    ha = recordio.IRHeader(0, [1.0, 2.0, 3.0], 5, 0)
    hdr, payload = recordio.unpack(recordio.pack(ha, b"xyz"))
    assert hdr.flag == 3 and np.allclose(hdr.label, [1, 2, 3]) and payload == b"xyz"

    # sequential file
    p = str(tmp_path / "seq.rec")
    with recordio.MXRecordIO(p, "w") as w:
        for i in range(5):
            w.write(recordio.pack(recordio.IRHeader(0, float(i), i, 0),
                                  bytes([i]) * 4))
    r = recordio.MXRecordIO(p, "r")
    labels = []
    while True:
        rec = r.read()
        if rec is None:
            break
        hdr, payload = recordio.unpack(rec)
        labels.append(hdr.label)
        assert payload == bytes([int(hdr.label)]) * 4
    assert labels == [0.0, 1.0, 2.0, 3.0, 4.0]
    r.reset()
    assert recordio.unpack(r.read())[0].label == 0.0

    # indexed file
    pi = str(tmp_path / "idx.rec")
    ix = str(tmp_path / "idx.idx")
    w = recordio.MXIndexedRecordIO(ix, pi, "w")
    for i in range(4):
        w.write_idx(10 + i, recordio.pack(recordio.IRHeader(0, float(i), i, 0),
                                          b"p%d" % i))
    w.close()
    r = recordio.MXIndexedRecordIO(ix, pi, "r")
    assert sorted(r.keys()) == [10, 11, 12, 13]
    hdr, payload = recordio.unpack(r.read_idx(12))
    assert hdr.label == 2.0 and payload == b"p2"
