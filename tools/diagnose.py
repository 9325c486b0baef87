#!/usr/bin/env python3
"""Environment diagnosis (reference tools/diagnose.py): print the platform,
ROCm/torch/dtmx versions, GPU inventory and the dtmx/cluster environment
variables that shape a run — the first thing to attach to a bug report."""
import os
import platform
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    print("----------Python Info----------")
    print("Version      :", platform.python_version())
    print("Arch         :", platform.machine())
    print("Platform     :", platform.platform())

    print("----------dtmx Info----------")
    try:
        import dtmx

        print("dtmx         :", os.path.dirname(dtmx.__file__))
        from dtmx.ops.hip import get_ext

        ext = get_ext()
        print("native _C    :", "loaded" if ext is not None else "NOT BUILT")
    except Exception as e:
        print("dtmx import FAILED:", e)

    print("----------Torch/ROCm Info----------")
    try:
        import torch

        print("torch        :", torch.__version__)
        print("hip          :", getattr(torch.version, "hip", None))
        print("cuda_avail   :", torch.cuda.is_available())
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                p = torch.cuda.get_device_properties(i)
                print(f"gpu[{i}]       : {p.name} gcnArch={getattr(p, 'gcnArchName', '?')} "
                      f"{p.total_memory / 2**30:.0f} GiB")
        import torch.distributed as dist

        print("dist backends:", [b for b in ("nccl", "gloo")
                                 if getattr(dist, f"is_{b}_available")()])
    except Exception as e:
        print("torch probe FAILED:", e)

    print("----------Environment----------")
    prefixes = ("DTMX_", "DMLC_", "PS_", "ELASTIC", "WORKER_HOST_FILE",
                "NEW_WORKER", "EPOCH_BEGIN", "RANK", "WORLD_SIZE", "MASTER_",
                "LOCAL_RANK", "HSA_", "NCCL_", "RCCL_", "PYTORCH_ROCM_ARCH")
    for k in sorted(os.environ):
        if k.startswith(prefixes):
            print(f"{k}={os.environ[k]}")


if __name__ == "__main__":
    main()
