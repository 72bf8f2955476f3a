"""Distributed helpers.

Two transports:
 * CPU path: an in-place allreduce callback backed by torch.distributed
   (gloo) — used by multi-process CPU runs and tests.
 * GPU path: RCCL inside the native engine.  torch.distributed (any backend)
   is used only to broadcast the 128-byte ncclUniqueId at startup; after
   that no Python sits in the communication path.
"""
import numpy as np


def gloo_allreduce_callback():
    import torch
    import torch.distributed as dist

    def allreduce(arr: np.ndarray, op: str = "sum"):
        t = torch.from_numpy(arr)
        dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max"
                        else dist.ReduceOp.SUM)
    return allreduce


def broadcast_rccl_id(rank):
    """Rank 0 creates the RCCL unique id; broadcast via torch.distributed
    (must already be initialised, any backend)."""
    import torch
    import torch.distributed as dist
    from . import _core
    if rank == 0:
        rid = _core.rccl_unique_id()
        buf = torch.tensor(bytearray(rid), dtype=torch.uint8)
    else:
        buf = torch.zeros(128, dtype=torch.uint8)
    dist.broadcast(buf, src=0)
    return bytes(buf.tolist())
