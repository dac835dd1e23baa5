"""Two-process RCCL bootstrap probe (runs under torchrun on one node).

Validates the file-based ncclUniqueId rendezvous of da_init plus a
scalar allreduce.  On a 1-GPU box both ranks share device 0 (RCCL may
refuse duplicate devices — reaching that error still proves the
rendezvous works); on an 8-GPU node each rank gets its own device via
LOCAL_RANK and the allreduce must return the true sum.
"""
import ctypes
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import distributedarrays_jl_amd as dja
from distributedarrays_jl_amd._ffi import lib, check


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    ngpu = 1
    try:
        import torch
        ngpu = max(1, torch.cuda.device_count())
    except Exception:
        pass
    dev = int(os.environ.get("LOCAL_RANK", str(rank))) % ngpu
    try:
        dja.comm.init(device=dev)
        buf = ctypes.c_double(1.0 + rank)
        check(lib.da_allreduce(ctypes.byref(buf), 1, 0, 0))
        expect = world * (world + 1) / 2.0
        ok = abs(buf.value - expect) < 1e-12
        print("rank %d dev %d: allreduce=%r expect=%r ok=%r"
              % (rank, dev, buf.value, expect, ok), flush=True)
        sys.exit(0 if ok else 1)
    except Exception as e:
        print("rank %d dev %d: EXC %s" % (rank, dev, repr(e)[:300]),
              flush=True)
        sys.exit(2)


if __name__ == "__main__":
    main()
