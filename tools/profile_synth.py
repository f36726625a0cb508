#!/usr/bin/env python3
"""torch.profiler (with stacks) over a few synthetic-small steps — identifies
the python call sites behind the remaining at::native glue kernels."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import distributed_embeddings_amd as de
from distributed_embeddings_amd.models.config import synthetic_models
from distributed_embeddings_amd.models.synthetic import SyntheticModel, expand_tables
from distributed_embeddings_amd.parallel.optim import SparseEmbeddingOptimizer
from distributed_embeddings_amd.utils.input_gen import make_batch


def main():
    torch.manual_seed(0)
    cfg = synthetic_models["small"]
    with torch.device("cuda"):
        model = SyntheticModel(cfg, strategy="memory_balanced")
    tables, imap, hotness = expand_tables(cfg)
    sizes = [tables[t][0] for t in imap]
    b = 65536
    g = torch.Generator().manual_seed(1)
    cats = [c.cuda() for c in make_batch(sizes, hotness, b, 1.05, generator=g,
                                         keep_hot_dim=True)]
    num = torch.rand(b, cfg.num_numerical_features, device="cuda")
    labels = torch.randint(0, 2, (b, 1), device="cuda").float()
    opt = SparseEmbeddingOptimizer(model.parameters(), lr=1e-3, method="sgd")
    model.embeddings.enable_fused_sgd(1e-3)
    lossf = torch.nn.BCEWithLogitsLoss()

    def step():
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(num, cats)
            loss = lossf(out.float(), labels)
        loss.backward()
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 with_stack=False) as prof:
        for _ in range(3):
            step()
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=False).table(
        sort_by="cuda_time_total", row_limit=30, max_name_column_width=70))


if __name__ == "__main__":
    main()
