#!/usr/bin/env python3
"""Bench-shaped lookup/update kernel workload for PMC bandwidth capture.

Runs the DLRM-bs64k-shaped hot kernels a FIXED number of times and prints the
theoretical HBM byte counts, so a `rocprofv3 --pmc TCC_EA_RDREQ...` run over
this script yields measured-vs-theoretical bandwidth.

  gpurun: rocprofv3 --pmc <RD> <WR> -d gpurun_out/pmcbw -- python tools/pmc_bw.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

ITERS = 10
VOCAB = 10_000_000   # 5.1 GB fp32 table (bench tables total 188M rows)
WIDTH = 128
NNZ = 1_700_000      # 26 tables x 65536 samples


def main():
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()
    torch.manual_seed(0)
    w = torch.randn(VOCAB, WIDTH, device="cuda")
    ids = torch.randint(0, VOCAB, (NNZ,), device="cuda")
    splits = torch.arange(0, NNZ + 1, 1, device="cuda")
    lr = torch.tensor([1e-3], device="cuda")
    grad16 = torch.randn(NNZ, WIDTH, device="cuda").bfloat16()
    state = torch.empty(0, device="cuda")

    # warmup
    out = ext.csr_lookup_forward(w, ids, splits, False, True)
    ext.csr_fused_optimizer_apply(w, state, ids, splits, grad16, lr,
                                  False, False, 0.0)
    torch.cuda.synchronize()

    for _ in range(ITERS):
        out = ext.csr_lookup_forward(w, ids, splits, False, True)
    torch.cuda.synchronize()
    for _ in range(ITERS):
        ext.csr_fused_optimizer_apply(w, state, ids, splits, grad16, lr,
                                      False, False, 0.0)
    torch.cuda.synchronize()

    fwd_rd = ITERS * (NNZ * WIDTH * 4 + NNZ * 8 * 2)      # param rows + ids/splits
    fwd_wr = ITERS * (NNZ * WIDTH * 2)                     # bf16 out
    # fused update: sort(4 passes rd+wr of 8B keys) + expand + seg-sum reads
    # grad rows (bf16) + read-modify-write unique param rows (fp32)
    upd_sort = ITERS * (NNZ * 8 * 2 * 4 * 2)
    upd_core = ITERS * (NNZ * WIDTH * 2 + NNZ * WIDTH * 4 * 2 // 1)
    print(f"forward: iters={ITERS} theoretical rd={fwd_rd/1e9:.2f} GB "
          f"wr={fwd_wr/1e9:.2f} GB")
    print(f"update:  iters={ITERS} sort~{upd_sort/1e9:.2f} GB "
          f"core~{upd_core/1e9:.2f} GB (upper bound: all ids unique)")
    print("pmc_bw done", float(out.float().sum()))


if __name__ == "__main__":
    main()
