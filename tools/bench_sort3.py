#!/usr/bin/env python3
"""Sweep the radix-sort tile variants (DE_SORT_VARIANT) + rocPRIM baseline
on the full sparse backward, at DLRM-backward and fused-group sizes.

Run on GPU: python tools/bench_sort3.py       (spawns one subprocess/variant)
"""
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

CASES = [(213_000, 188_000_000, "dlrm-26tbl"), (1_700_000, 188_000_000, "dlrm-bs64k"),
         (2_000_000, 4_000_000, "small-grp")]


def worker():
    import time
    import torch
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()

    def timeit(fn, iters=30):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    torch.manual_seed(0)
    out = []
    for nnz, vocab, label in CASES:
        rows = max(nnz // 22, 1)
        gr = torch.randn(rows, 64, device="cuda")
        ids = torch.randint(0, vocab, (nnz,), device="cuda")
        splits = torch.linspace(0, nnz, rows + 1, device="cuda").long()
        us = timeit(lambda: ext.csr_lookup_backward(gr, ids, splits, vocab, False))
        out.append(f"{label}: {us:.0f} us")
    print(" | ".join(out), flush=True)


def main():
    if os.environ.get("_SORT_WORKER"):
        worker()
        return
    env = dict(os.environ, _SORT_WORKER="1")
    for v in range(8):
        e = dict(env, DE_SORT_VARIANT=str(v))
        print(f"variant {v}: ", end="", flush=True)
        subprocess.run([sys.executable, __file__], env=e, check=True)
    e = dict(env, DE_USE_ROCPRIM_SORT="1")
    print("rocprim  : ", end="", flush=True)
    subprocess.run([sys.executable, __file__], env=e, check=True)


if __name__ == "__main__":
    main()
