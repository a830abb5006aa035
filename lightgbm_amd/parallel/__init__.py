"""Distributed helpers: wire torch.distributed collectives into the native Network seam.

Replaces the reference's hand-rolled TCP socket mesh (src/network/linkers_socket.cpp) for
multi-process CPU training and provides the bootstrap used by the multi-GPU bench
(one process per GPU; RCCL over xGMI runs natively inside the HIP learner, the gloo
store here is only used for rendezvous / bin-mapper broadcast).
"""
import ctypes
import os

import numpy as np

from ..basic import _LIB, _safe_call

__all__ = ["init_network_from_torch_distributed", "free_network", "init_process_group"]

_ALLGATHER_CB = None  # keep callback alive


def init_process_group(backend="gloo"):
    """Initialize torch.distributed from torchrun env vars; no-op if already done."""
    import torch.distributed as dist
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ:
        return 0, 1
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    import datetime
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=300))
    return dist.get_rank(), dist.get_world_size()


def init_network_from_torch_distributed(group=None):
    """Register a torch.distributed(gloo/nccl)-backed allgather with the native Network.

    The native signature (reference LGBM_NetworkInitWithFunctions AllgatherExtFunction):
    allgather(char* input, int input_size, const int* block_start, const int* block_len,
              int num_block, char* output, int output_size).
    """
    global _ALLGATHER_CB
    import torch
    import torch.distributed as dist
    if not dist.is_initialized():
        init_process_group()
    if not dist.is_initialized():
        return  # single process
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)

    cb_type = ctypes.CFUNCTYPE(None, ctypes.POINTER(ctypes.c_char), ctypes.c_int,
                               ctypes.POINTER(ctypes.c_int), ctypes.POINTER(ctypes.c_int),
                               ctypes.c_int, ctypes.POINTER(ctypes.c_char), ctypes.c_int)

    def _allgather(inp, input_size, block_start, block_len, num_block, out, output_size):
        buf = np.ctypeslib.as_array(
            ctypes.cast(inp, ctypes.POINTER(ctypes.c_uint8)), shape=(input_size,))
        t = torch.from_numpy(buf.copy())
        gathered = [torch.empty(input_size, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(gathered, t, group=group)
        outbuf = np.ctypeslib.as_array(
            ctypes.cast(out, ctypes.POINTER(ctypes.c_uint8)), shape=(output_size,))
        for r in range(world):
            s = block_start[r]
            n = block_len[r]
            outbuf[s:s + n] = gathered[r].numpy()[:n]

    _ALLGATHER_CB = cb_type(_allgather)
    _safe_call(_LIB.LGBM_NetworkInitWithFunctions(
        ctypes.c_int(world), ctypes.c_int(rank), None,
        ctypes.cast(_ALLGATHER_CB, ctypes.c_void_p)))


def free_network():
    global _ALLGATHER_CB
    _safe_call(_LIB.LGBM_NetworkFree())
    _ALLGATHER_CB = None
