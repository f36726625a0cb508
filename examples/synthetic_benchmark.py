#!/usr/bin/env python3
"""Synthetic-model benchmark harness.

Capability parity with the reference
``examples/benchmarks/synthetic_models/main.py``: picks one of the seven
model scales (tiny ... colossal), builds ``SyntheticModel`` (per-table
Embedding(combiner='sum') + DistributedEmbedding(memory_balanced) + avg-pool
interaction + MLP), generates power-law inputs, runs warmup then a timed
loop with a loss-allreduce sync, and prints ms/iteration.

Launch: torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N \
    examples/synthetic_benchmark.py --model small
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import distributed_embeddings_amd as de
from distributed_embeddings_amd.models.config import synthetic_models
from distributed_embeddings_amd.models.synthetic import SyntheticModel, expand_tables
from distributed_embeddings_amd.parallel.optim import SparseEmbeddingOptimizer
from distributed_embeddings_amd.utils.input_gen import make_batch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="tiny", choices=list(synthetic_models))
    p.add_argument("--batch-size", type=int, default=65536, help="global batch")
    p.add_argument("--num-steps", type=int, default=100)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--alpha", type=float, default=1.05)
    p.add_argument("--column-slice-threshold", type=int, default=None)
    p.add_argument("--dp-input", action="store_true", default=True)
    p.add_argument("--optimizer", default="adagrad", choices=["adagrad", "sgd"])
    p.add_argument("--pool", type=int, default=4)
    p.add_argument("--graph", dest="graph", action="store_true", default=True,
                   help="hipGraph-capture the step (world==1; eager fallback)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    return p.parse_args()


def maybe_enable_tunableop():
    try:
        import torch.cuda.tunable as tunable
        csv = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                           "profiles", "tunableop_gfx950.csv")
        if os.path.exists(csv) and torch.cuda.is_available():
            tunable.enable(True)
            tunable.tuning_enable(False)
            tunable.read_file(csv)
    except Exception:
        pass


def main():
    args = parse_args()
    maybe_enable_tunableop()
    if "RANK" in os.environ and int(os.environ.get("WORLD_SIZE", "1")) > 1:
        torch.distributed.init_process_group(
            "nccl" if torch.cuda.is_available() else "gloo")
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
    rank, world = de.comm.rank(), de.comm.world_size()
    device = torch.device("cuda", torch.cuda.current_device()) \
        if torch.cuda.is_available() else torch.device("cpu")

    cfg = synthetic_models[args.model]
    with torch.device(device):
        model = SyntheticModel(cfg, column_slice_threshold=args.column_slice_threshold)
    tables, input_map, hotness = expand_tables(cfg)
    table_sizes = [tables[t][0] for t in input_map]
    local_bs = args.batch_size // world

    # cats flat-packed per entry: graph replay needs ONE device copy
    pool = []
    cat_shapes = None
    for i in range(args.pool):
        g = torch.Generator().manual_seed(23 + 131 * rank + i)
        cats = [c.to(device) for c in
                make_batch(table_sizes, hotness, local_bs, args.alpha,
                           generator=g, keep_hot_dim=True)]
        if cat_shapes is None:
            cat_shapes = [c.shape for c in cats]
            cat_sizes = [c.numel() for c in cats]
        flat = torch.cat([c.reshape(-1) for c in cats])
        num = torch.rand(local_bs, cfg.num_numerical_features, device=device)
        labels = torch.randint(0, 2, (local_bs, 1), device=device).float()
        pool.append((num, flat, labels))

    def carve(flat):
        return [p.view(s) for p, s in
                zip(torch.split(flat, cat_sizes), cat_shapes)]

    opt = de.DistributedOptimizer(SparseEmbeddingOptimizer(
        model.parameters(), lr=0.01, method=args.optimizer), average=False)
    model.embeddings.enable_fused_optimizer(args.optimizer, 0.01)
    de.broadcast_parameters(model)
    _loss_sum = torch.nn.BCEWithLogitsLoss(reduction="sum")
    loss_fn = lambda lg, lb: _loss_sum(lg, lb) / args.batch_size

    def run_step(num, cats, labels, set_to_none=True):
        opt.zero_grad(set_to_none=set_to_none)
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=device.type == "cuda"):
            loss = loss_fn(model(num, cats).float(), labels)
        loss.backward()
        opt.step()
        return loss

    graph = None
    if args.graph and world == 1 and device.type == "cuda":
        s_num, s_flat, s_labels = (t.clone() for t in pool[0])
        s_cats = carve(s_flat)
        try:
            for i in range(max(args.warmup, 2)):
                run_step(s_num, s_cats, s_labels, set_to_none=False)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                run_step(s_num, s_cats, s_labels, set_to_none=False)
            graph = g
        except Exception as e:
            print(f"# graph capture failed ({type(e).__name__}); eager")
            graph = None

    def step(i):
        num, flat, labels = pool[i % len(pool)]
        if graph is not None:
            s_num.copy_(num)
            s_flat.copy_(flat)
            s_labels.copy_(labels)
            graph.replay()
            return torch.zeros(1)
        return run_step(num, carve(flat), labels)

    for i in range(args.warmup):
        step(i)
    loss = step(0)
    lt = torch.tensor([float(loss.detach())])
    de.comm.allreduce_sum_(lt)  # sync + flush (parity: reference main.py:140-158)

    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    for i in range(args.num_steps):
        loss = step(i)
    lt = torch.tensor([float(loss.detach())])
    de.comm.allreduce_sum_(lt)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.time() - t0

    if rank == 0:
        ms = dt / args.num_steps * 1000
        print(f"model={cfg.name} world={world} global_bs={args.batch_size} "
              f"{ms:.3f} ms/iteration ({args.batch_size / ms * 1000:.0f} samples/s)")


if __name__ == "__main__":
    main()
