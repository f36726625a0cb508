#!/usr/bin/env python3
"""Micro-benchmark: interaction kernels, old [B,F,D] vs packed layouts."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

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


def sweep_wpb():
    import subprocess
    env = dict(os.environ, _DI_WORKER="1")
    for w in (2, 4, 8):
        e = dict(env, DE_DI_WPB=str(w))
        print(f"--- WPB={w} ---", flush=True)
        subprocess.run([sys.executable, __file__], env=e, check=True)


def main():
    if "_DI_WORKER" not in os.environ and os.environ.get("DI_SWEEP") == "1":
        sweep_wpb()
        return
    torch.manual_seed(0)
    B, P, D = 65536, 26, 128
    F = P + 1
    out_w = 512
    feats = torch.randn(B, F, D, device="cuda").bfloat16()
    packed_f = torch.randn(P, B, D, device="cuda").bfloat16()
    bottom = torch.randn(B, D, device="cuda").bfloat16()
    perm = torch.arange(P, dtype=torch.int32, device="cuda")
    gout = torch.randn(B, out_w, device="cuda").bfloat16()

    t = timeit(lambda: ext.dot_interact_fwd(feats, out_w))
    print(f"fwd  old [B,F,D]        : {t:7.1f} us")
    t = timeit(lambda: ext.dot_interact_fwd_packed(bottom, packed_f, perm,
                                                   out_w, False))
    print(f"fwd  packed feat-major  : {t:7.1f} us")
    t = timeit(lambda: ext.dot_interact_bwd(gout, feats))
    print(f"bwd  old [B,F,D]        : {t:7.1f} us")
    t = timeit(lambda: ext.dot_interact_bwd_packed(gout, bottom, packed_f,
                                                   perm, False))
    print(f"bwd  packed feat-major  : {t:7.1f} us")


if __name__ == "__main__":
    main()
