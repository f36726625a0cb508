#!/usr/bin/env python3
"""Probe: can RCCL run a 2-rank communicator with both ranks on cuda:0?

Run on a 1-GPU box: python tools/probe_rccl2.py
Prints the per-rank outcome (and full tracebacks to stderr).
"""
import os
import sys
import traceback


def entry(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch
    import torch.distributed as dist
    try:
        torch.cuda.set_device(0)
        dist.init_process_group("nccl", rank=rank, world_size=world)
        t = torch.ones(4, device="cuda") * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        print(f"[rank {rank}] all_reduce ok: {t.tolist()}", flush=True)
        # all_to_all_single with uneven splits (the path the library leans on)
        send = torch.arange(6, device="cuda", dtype=torch.float32)
        out_splits = [2, 4] if rank == 0 else [4, 2]
        in_splits = [2, 4] if rank == 0 else [4, 2]
        recv = torch.empty(6, device="cuda")
        dist.all_to_all_single(recv, send, out_splits, in_splits)
        torch.cuda.synchronize()
        print(f"[rank {rank}] a2a ok: {recv.tolist()}", flush=True)
        dist.destroy_process_group()
    except Exception:
        traceback.print_exc()
        sys.exit(1)


def main():
    import socket
    import torch.multiprocessing as mp
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=entry, args=(r, 2, port)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    codes = [p.exitcode for p in procs]
    print("exit codes:", codes, flush=True)
    sys.exit(0 if codes == [0, 0] else 1)


if __name__ == "__main__":
    main()
