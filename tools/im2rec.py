#!/usr/bin/env python3
"""Pack arrays into a RecordIO file (reference tools/im2rec.py / im2rec.cc).

Records carry either JPEG payloads (libjpeg encode; --jpeg) or raw uint8
HWC behind the reference IRHeader (flag,label,id,id2):

    python tools/im2rec.py out.rec --from-npz data.npz          # x:[N,H,W,C] u8, y:[N]
    python tools/im2rec.py out.rec --synthetic N H W C CLASSES  # random data
    python tools/im2rec.py out.rec --synthetic ... --jpeg --quality 90
"""
import argparse
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def pack_record(label: float, raw: bytes, idx: int) -> bytes:
    header = struct.pack("<IfQQ", 0, float(label), idx, 0)
    return header + raw


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("out")
    ap.add_argument("--from-npz", type=str, default=None)
    ap.add_argument("--synthetic", type=int, nargs=5, default=None,
                    metavar=("N", "H", "W", "C", "CLASSES"))
    ap.add_argument("--jpeg", action="store_true",
                    help="JPEG-encode payloads (reference im2rec default)")
    ap.add_argument("--quality", type=int, default=95)
    args = ap.parse_args()
    from dtmx.ops.hip import require_ext

    ext = require_ext()
    if args.from_npz:
        blob = np.load(args.from_npz)
        x, y = blob["x"].astype(np.uint8), blob["y"]
    else:
        n, h, w, c, k = args.synthetic or (256, 32, 32, 3, 10)
        rng = np.random.RandomState(0)
        x = rng.randint(0, 256, (n, h, w, c), dtype=np.uint8)
        y = rng.randint(0, k, (n,))
    if args.jpeg:
        h, w, c = x.shape[1:]
        records = [pack_record(y[i], ext.encode_jpeg(x[i].tobytes(), h, w, c,
                                                     args.quality), i)
                   for i in range(len(x))]
    else:
        records = [pack_record(y[i], x[i].tobytes(), i) for i in range(len(x))]
    ext.write_recordio(args.out, records)
    print(f"wrote {len(records)} records to {args.out}")


if __name__ == "__main__":
    main()
