#!/usr/bin/env python3
"""xGMI all-to-all / allreduce bandwidth probe (round-2 prep).

Launch: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N \
    tools/bench_a2a.py [--min-mb 1] [--max-mb 256]

Prints per-message-size busbw for all_to_all_single, all_gather and
reduce_scatter — the three collectives on DistributedEmbedding's hot path.
On an 8-GPU xGMI full mesh the a2a should approach the 7-link aggregate
(~1 TB/s per GPU); ring allreduce is per-link bound (~150 GB/s per
direction).  Works on CPU/gloo too (numbers meaningless, plumbing check).
"""

import argparse
import os
import time

import torch
import torch.distributed as dist


def bench(fn, iters=20, warmup=5, device=None):
    for _ in range(warmup):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dist.barrier()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--min-mb", type=float, default=1)
    p.add_argument("--max-mb", type=float, default=256)
    args = p.parse_args()

    have_gpu = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if have_gpu:
        torch.cuda.set_device(local_rank)
    dist.init_process_group("nccl" if have_gpu else "gloo")
    rank, world = dist.get_rank(), dist.get_world_size()
    device = torch.device("cuda", local_rank) if have_gpu else "cpu"

    mb = args.min_mb
    while mb <= args.max_mb:
        n = int(mb * 1e6 / 2)  # bf16 elements
        dt = torch.bfloat16 if have_gpu else torch.float32
        x = torch.randn(n, dtype=dt, device=device)
        y = torch.empty_like(x)

        t_a2a = bench(lambda: dist.all_to_all_single(y, x))
        g = torch.empty(n * world, dtype=dt, device=device)
        t_ag = bench(lambda: dist.all_gather_into_tensor(g, x))
        r = torch.empty(n // world, dtype=dt, device=device)
        xs = x[: (n // world) * world]
        t_rs = bench(lambda: dist.reduce_scatter_tensor(r, xs))

        bytes_ = n * x.element_size()
        if rank == 0:
            # busbw conventions: a2a moves (W-1)/W of the buffer off-GPU
            f = (world - 1) / world
            print(f"{mb:8.1f} MB  a2a {bytes_ * f / t_a2a / 1e9:8.1f} GB/s   "
                  f"ag {bytes_ * (world - 1) / t_ag / 1e9:8.1f} GB/s   "
                  f"rs {bytes_ * f / t_rs / 1e9:8.1f} GB/s")
        mb *= 4


if __name__ == "__main__":
    main()
