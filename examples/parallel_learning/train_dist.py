"""Distributed training example: one process per GPU (or per CPU rank).

Launch:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 train_dist.py [--device cuda]
"""
import argparse
import ctypes
import os
import sys

# allow running from a repo checkout without installing the package
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import numpy as np
import torch.distributed as dist

import lightgbm_amd as lgb
from lightgbm_amd.basic import _LIB
from lightgbm_amd.parallel import init_network_from_torch_distributed


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cpu", choices=["cpu", "cuda"])
    ap.add_argument("--rows", type=int, default=200_000)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        dist.init_process_group("gloo")
        init_network_from_torch_distributed()
    use_gpu = args.device == "cuda"
    if use_gpu:
        _LIB.LGBM_GPUSetDevice(ctypes.c_int(local_rank))
        if world > 1:
            buf = ctypes.create_string_buffer(256)
            size = ctypes.c_int(0)
            if rank == 0:
                assert _LIB.LGBM_GPUGetUniqueId(buf, ctypes.byref(size)) == 0
                payload = [bytes(buf.raw[: size.value])]
            else:
                payload = [None]
            dist.broadcast_object_list(payload, src=0)
            assert _LIB.LGBM_GPUNetworkInit(ctypes.c_int(world), ctypes.c_int(rank),
                                            payload[0]) == 0

    # every rank: same binning reference (common seed), own shard
    rng = np.random.RandomState(1234)
    Xr = rng.rand(50_000, 20)
    yr = (Xr[:, 0] + Xr[:, 1] > 1.0).astype(np.float32)
    ref = lgb.Dataset(Xr, label=yr).construct()
    shard_rng = np.random.RandomState(1000 + rank)
    X = shard_rng.rand(args.rows // max(world, 1), 20)
    y = (X[:, 0] + X[:, 1] > 1.0).astype(np.float32)
    train = ref.create_valid(X, label=y)

    params = {"objective": "binary", "tree_learner": "data", "num_leaves": 63,
              "verbosity": -1}
    if use_gpu:
        params["device_type"] = "cuda"
    bst = lgb.train(params, train, 50)
    if rank == 0:
        print(f"trained {bst.num_trees()} trees on {world} rank(s); "
              f"train acc = {((bst.predict(X) > 0.5) == y).mean():.4f}")
    if world > 1:
        if use_gpu:
            _LIB.LGBM_GPUNetworkFree()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
