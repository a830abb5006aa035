"""Worker for the multi-rank GPU data-parallel tests.

Launched under torchrun with N ranks that all share GPU 0 (single-GPU box).
RCCL refuses duplicate devices, so the cross-rank transport here is the host
seam (migbm::Network over gloo) — the DEVICE code path (histogram reduce,
owned-feature gain scan, winner sync, global leaf counts) is identical to the
RCCL path; only the reduction transport differs (GpuComm::HostBounce).

Writes model_rank{r}.txt per rank and result.json from rank 0 into --out-dir.
Env: MIGBM_DIST_HIST selects allreduce | reduce_scatter | auto.
"""
import argparse
import json
import os
import sys
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parents[2]
sys.path.insert(0, str(REPO))


def make_data(n, d, seed):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, d).astype(np.float32)
    logit = 1.2 * X[:, 0] - 0.8 * X[:, 1] + 0.9 * X[:, 2] * X[:, 3] + 0.5 * X[:, 4]
    y = (logit + 1.0 * rng.randn(n) > 0).astype(np.float32)
    return X, y


def auc(y, p):
    order = np.argsort(-p, kind="stable")
    ys = y[order]
    n_pos = ys.sum()
    n_neg = len(ys) - n_pos
    ranks = np.arange(1, len(ys) + 1)
    return 1.0 - (ranks[ys > 0].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out-dir", required=True)
    ap.add_argument("--rows", type=int, default=120_000)
    ap.add_argument("--features", type=int, default=20)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--num-leaves", type=int, default=31)
    ap.add_argument("--max-bin", type=int, default=63)
    args = ap.parse_args()

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])

    import torch.distributed as dist
    import lightgbm_amd as lgb
    from lightgbm_amd.parallel import (init_network_from_torch_distributed,
                                       init_process_group, free_network)

    init_process_group("gloo")
    init_network_from_torch_distributed()

    # identical bin mappers everywhere: common-seed reference sample, then each
    # rank bins its own row shard against it
    X, y = make_data(args.rows, args.features, seed=7)
    ref = lgb.Dataset(X[:50_000], label=y[:50_000],
                      params={"max_bin": args.max_bin}).construct()
    shard = slice(rank * args.rows // world, (rank + 1) * args.rows // world)
    train = ref.create_valid(X[shard], label=y[shard]).construct()

    params = {
        "objective": "binary",
        "device_type": "cuda",
        "tree_learner": "data",
        "max_bin": args.max_bin,
        "num_leaves": args.num_leaves,
        "min_data_in_leaf": 1,
        "min_sum_hessian_in_leaf": 5,
        "verbosity": 0,
    }
    bst = lgb.Booster(params=params, train_set=train)
    for _ in range(args.iters):
        bst.update()

    out = Path(args.out_dir)
    model = bst.model_to_string()
    (out / f"model_rank{rank}.txt").write_text(model)
    dist.barrier()

    if rank == 0:
        p = bst.predict(X)
        res = {"auc_full": float(auc(y, p)), "world": world,
               "num_trees": bst.num_trees(),
               "mode": os.environ.get("MIGBM_DIST_HIST", "auto")}
        (out / "result.json").write_text(json.dumps(res))
    dist.barrier()
    free_network()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
