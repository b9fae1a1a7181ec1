#!/usr/bin/env python3
"""DDP training-step workload (BASELINE config #4).

Runs as a gang task (one rank per GPU): a synthetic MLP trains with the
framework's bucketed all-reduce over RCCL/xGMI — the
TensorFlow-Distributed-recipe analogue.  When SHIPYARD_TASK_INPUT_DIR
contains SYSHARD shards, they are staged NVMe->HBM with GPU
decode+verify first (the HIP data-mover path).

Usage (inside a multi_instance task, or under torchrun):
    python benchmarks/ddp_step.py [--steps N] [--hidden H] [--layers L]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--hidden", type=int, default=4096)
    p.add_argument("--layers", type=int, default=8)
    p.add_argument("--batch", type=int, default=64)
    args = p.parse_args()

    import torch

    from shipyard_amd.comm import GangComm

    comm = GangComm()
    dev = comm.device
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32

    # optional: stage shards through the HIP data mover
    staged_bytes = 0
    in_dir = os.environ.get("SHIPYARD_TASK_INPUT_DIR")
    if in_dir and dev.type == "cuda":
        from shipyard_amd.data.stager import ShardStager

        shards = sorted(Path(in_dir).glob("*.syshard"))
        if shards:
            stager = ShardStager(device=dev)
            for res in stager.stage_many(shards).values():
                staged_bytes += res.raw_bytes

    H, L, B = args.hidden, args.layers, args.batch
    layers = []
    for _ in range(L):
        layers += [torch.nn.Linear(H, H), torch.nn.GELU()]
    model = torch.nn.Sequential(*layers).to(dev, dtype)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    x = torch.randn(B, H, device=dev, dtype=dtype)
    y = torch.randn(B, H, device=dev, dtype=dtype)

    def step():
        opt.zero_grad(set_to_none=False)
        loss = torch.nn.functional.mse_loss(model(x).float(), y.float())
        loss.backward()
        comm.all_reduce_bucketed_(
            [q.grad for q in model.parameters() if q.grad is not None])
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    comm.barrier()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    elapsed = comm.max_scalar(time.perf_counter() - t0)
    comm.barrier()

    if comm.rank == 0:
        n_params = sum(q.numel() for q in model.parameters())
        print(json.dumps({
            "workload": "ddp-step",
            "world": comm.world,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "params": n_params,
            "grad_bytes_allreduced_per_step": n_params * 2,
            "staged_shard_bytes": staged_bytes,
            "loss": float(loss.item()),
        }), flush=True)
    comm.shutdown()
    return 0


if __name__ == "__main__":
    sys.exit(main())
