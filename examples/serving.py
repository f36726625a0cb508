#!/usr/bin/env python3
"""DLRM serving-latency demo: load checkpointed tables, score candidate
batches under ``torch.inference_mode`` and report p50/p95/p99 latency.

The full Criteo-1TB-class model fits ONE MI355X (96 GB fp32 tables — or
48 GB with ``--table-dtype bf16``), so single-GPU serving needs no sharding;
``--graph`` captures the scoring step in a hipGraph for launch-bound small
batches.

  python examples/serving.py --batch-size 4096 --iters 200 [--graph]
  python examples/serving.py --weights dump.npz        # from dlrm_main --dump-embeddings
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from distributed_embeddings_amd.models.config import CRITEO_1TB_TABLE_SIZES
from distributed_embeddings_amd.models.dlrm import DLRM


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--batch-size", type=int, default=4096,
                   help="candidates scored per request")
    p.add_argument("--iters", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--weights", default=None,
                   help="npz from dlrm_main --dump-embeddings, or a "
                        "checkpoint directory (table_*.npy; mmap-loaded)")
    p.add_argument("--table-dtype", default="fp32", choices=["fp32", "bf16"])
    p.add_argument("--table-size-cap", type=int, default=None)
    p.add_argument("--graph", action="store_true",
                   help="capture the scoring step in a hipGraph")
    return p.parse_args()


def main():
    args = parse_args()
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    sizes = CRITEO_1TB_TABLE_SIZES
    if args.table_size_cap:
        sizes = [min(s, args.table_size_cap) for s in sizes]
    tdt = torch.bfloat16 if args.table_dtype == "bf16" else torch.float32
    with torch.device(device):
        model = DLRM(sizes, embedding_dim=128, table_dtype=tdt)
    model.eval()
    if args.weights:
        if os.path.isdir(args.weights):
            from distributed_embeddings_amd import load_embedding_checkpoint
            load_embedding_checkpoint(model.embeddings, args.weights)
        else:
            with np.load(args.weights) as z:
                model.embeddings.set_weights([z[k] for k in z.files])

    b = args.batch_size
    g = torch.Generator().manual_seed(7)
    num = torch.rand(b, 13, generator=g).to(device)
    cats = [torch.randint(0, s, (b,), generator=g).to(device) for s in sizes]

    use_bf16 = device.type == "cuda"

    def score():
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16):
            return torch.sigmoid(model(num, cats).float())

    graph = None
    if args.graph and device.type == "cuda":
        with torch.inference_mode():
            for _ in range(3):
                out = score()
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                out = score()

    lat = []
    with torch.inference_mode():
        for i in range(args.warmup + args.iters):
            if device.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            if graph is not None:
                graph.replay()
            else:
                out = score()
            if device.type == "cuda":
                torch.cuda.synchronize()
            if i >= args.warmup:
                lat.append((time.perf_counter() - t0) * 1e3)
    lat.sort()
    p = lambda q: lat[min(len(lat) - 1, int(q * len(lat)))]
    print(f"batch {b}: p50 {p(0.50):.3f} ms  p95 {p(0.95):.3f} ms  "
          f"p99 {p(0.99):.3f} ms  ({b / p(0.50) * 1000:.0f} candidates/s)"
          f"{'  [hipGraph]' if graph is not None else ''}")
    # keep `out` alive for the graph's static buffers
    assert out.shape == (b, 1)


if __name__ == "__main__":
    main()
