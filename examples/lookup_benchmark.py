#!/usr/bin/env python3
"""Lookup micro-benchmark: ragged variable-hotness lookup fwd / grad / SGD
step vs the plain-PyTorch equivalent.

Capability parity with the reference ``examples/benchmarks/benchmark.py:23-98``
(voc=1e6, width=128, batch=16384, hotness<=500).
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--vocab", type=int, default=1_000_000)
    p.add_argument("--width", type=int, default=128)
    p.add_argument("--batch", type=int, default=16384)
    p.add_argument("--max-hotness", type=int, default=500)
    args = p.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    g = torch.Generator().manual_seed(0)
    lengths = torch.randint(1, args.max_hotness + 1, (args.batch,), generator=g)
    values = torch.randint(0, args.vocab, (int(lengths.sum()),), generator=g)
    ragged = Ragged.from_row_lengths(values.to(device), lengths.to(device))
    print(f"voc={args.vocab} width={args.width} batch={args.batch} "
          f"nnz={values.numel()} device={device}")

    emb = Embedding(args.vocab, args.width, combiner="sum").to(device)
    opt = SparseEmbeddingOptimizer(emb.parameters(), lr=0.01)

    print(f"custom fwd           {timeit(lambda: emb(ragged)):8.3f} ms")

    def fwd_bwd():
        opt.zero_grad()
        emb(ragged).sum().backward()
    print(f"custom fwd+grad      {timeit(fwd_bwd):8.3f} ms")

    def full_step():
        fwd_bwd()
        opt.step()
    print(f"custom fwd+grad+SGD  {timeit(full_step):8.3f} ms")

    # torch-native comparison: EmbeddingBag
    bag = torch.nn.EmbeddingBag(args.vocab, args.width, mode="sum",
                                sparse=True, include_last_offset=True).to(device)
    opt2 = torch.optim.SGD(bag.parameters(), lr=0.01)
    offsets = ragged.row_splits
    print(f"torch bag fwd        "
          f"{timeit(lambda: bag(ragged.values, offsets)):8.3f} ms")

    def bag_step():
        opt2.zero_grad()
        bag(ragged.values, offsets).sum().backward()
        opt2.step()
    print(f"torch bag fwd+g+SGD  {timeit(bag_step):8.3f} ms")


if __name__ == "__main__":
    main()
