"""One rank of a multi-process gloo training job (launched by
tests/test_multirank.py, one subprocess per rank).

Reproduces the engine path a Spark barrier task runs — SyncTrainer with
bucketed all-reduce over torch.distributed — at world sizes the local[2]
tests never reach (4, 8), and records everything the parent needs to check
the reference's implicit invariant (reference distributed.py:179-181,
255-261): identical final states on every rank, equal to a single-process
run on the concatenated data.

Writes to --outdir:
  state_<rank>.pt   final state_dict (CPU tensors)
  order_<rank>.json bucket all-reduce launch order per step
  losses_<rank>.json per-step loss
"""

from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn as nn


def build_model(seed: int = 99) -> nn.Module:
    torch.manual_seed(seed)
    # enough parameter tensors for several buckets at a small cap
    return nn.Sequential(
        nn.Linear(20, 64), nn.ReLU(),
        nn.Linear(64, 64), nn.ReLU(),
        nn.Linear(64, 32), nn.ReLU(),
        nn.Linear(32, 1),
    )


def shard(rank: int, n: int = 64):
    torch.manual_seed(1000 + rank)
    x = torch.randn(n, 20)
    y = (x.sum(dim=1, keepdim=True) > 0).float()
    return x, y


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rank", type=int, required=True)
    ap.add_argument("--world", type=int, required=True)
    ap.add_argument("--port", type=int, required=True)
    ap.add_argument("--outdir", type=str, required=True)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--bucket-cap-mb", type=float, default=0.02)
    args = ap.parse_args()

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(args.port)
    dist.init_process_group("gloo", rank=args.rank, world_size=args.world)

    from sparktorch_amd.parallel.sync import SyncTrainer

    model = build_model()
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    trainer = SyncTrainer(
        model, nn.MSELoss(), opt, device="cpu",
        world_size=args.world, bucket_cap_mb=args.bucket_cap_mb,
    )

    # record the bucket all-reduce launch order (must be identical on every
    # rank every step, or collectives would mismatch and corrupt grads)
    order: list = []
    fb = trainer.buckets
    orig_launch = fb._launch

    def traced_launch(bucket):
        order.append(bucket.index)
        orig_launch(bucket)

    fb._launch = traced_launch

    x, y = shard(args.rank)
    losses = [trainer.train_step(x, y) for _ in range(args.steps)]

    os.makedirs(args.outdir, exist_ok=True)
    torch.save(trainer.state_dict_cpu(), os.path.join(args.outdir, "state_%d.pt" % args.rank))
    with open(os.path.join(args.outdir, "order_%d.json" % args.rank), "w") as f:
        json.dump({"order": order, "n_buckets": len(fb.buckets)}, f)
    with open(os.path.join(args.outdir, "losses_%d.json" % args.rank), "w") as f:
        json.dump(losses, f)

    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
