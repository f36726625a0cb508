#!/usr/bin/env python3
"""DLRM training example (hybrid data+model parallel on MI355X).

Capability parity with the reference ``examples/dlrm/main.py``: full DLRM
training with DistributedEmbedding, custom train step with dp-grad averaging,
first-step parameter broadcast, AUC evaluation with allgathered predictions,
embedding dump via get_weights, warmup + polynomial-decay LR schedule, and a
binary Criteo dataset reader (or synthetic data when no dataset is given).

Launch (one process per GPU):
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      examples/dlrm_main.py --batch-size 65536 --num-batches 100
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import distributed_embeddings_amd as de
from distributed_embeddings_amd.models.config import CRITEO_1TB_TABLE_SIZES
from distributed_embeddings_amd.models.dlrm import DLRM
from distributed_embeddings_amd.parallel.optim import SparseEmbeddingOptimizer
from distributed_embeddings_amd.utils.criteo import RawBinaryDataset, SyntheticDLRMData
from distributed_embeddings_amd.utils.lr_schedule import WarmupPolyDecay


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--dataset-path", default=None,
                   help="path to split-binary Criteo (else synthetic)")
    p.add_argument("--learning-rate", type=float, default=24.0)
    p.add_argument("--batch-size", type=int, default=64 * 1024,
                   help="global batch size")
    p.add_argument("--num-batches", type=int, default=340)
    p.add_argument("--embedding-dim", type=int, default=128)
    p.add_argument("--dist-strategy", default="memory_balanced")
    p.add_argument("--dp-input", action="store_true")
    p.add_argument("--eval", action="store_true", help="run AUC eval at end")
    p.add_argument("--dump-embeddings", default=None,
                   help="npz path: dump full tables via get_weights")
    p.add_argument("--checkpoint-dir", default=None,
                   help="directory: save per-table .npy via "
                        "save_embedding_checkpoint (reshardable on load)")
    p.add_argument("--warmup-steps", type=int, default=8000)
    p.add_argument("--decay-start", type=int, default=70000)
    p.add_argument("--decay-steps", type=int, default=30000)
    p.add_argument("--table-size-cap", type=int, default=None,
                   help="cap per-table vocab (small-memory smoke runs)")
    p.add_argument("--learnable-labels", action="store_true",
                   help="synthetic labels = f(ids): loss must drop below "
                        "ln(2) if the optimizer scaling is right")
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--fused-optimizer", action="store_true",
                   help="in-backward fused SGD on the model-parallel tables "
                        "(no grad tensors, no host syncs; the schedule "
                        "updates the device-resident lr)")
    return p.parse_args()


def auc(scores: torch.Tensor, labels: torch.Tensor) -> float:
    """Rank-based AUC (parity: reference eval, examples/dlrm/main.py:223-243)."""
    order = scores.argsort()
    ranks = torch.empty_like(order, dtype=torch.float64)
    ranks[order] = torch.arange(1, len(scores) + 1, dtype=torch.float64)
    pos = labels.bool()
    n_pos = int(pos.sum())
    n_neg = len(labels) - n_pos
    if n_pos == 0 or n_neg == 0:
        return 0.5
    return float((ranks[pos].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg))


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
        torch.distributed.init_process_group("nccl" if torch.cuda.is_available() else "gloo")
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
    else:
        local_rank = 0
    rank, world = de.comm.rank(), de.comm.world_size()
    device = torch.device("cuda", local_rank) if torch.cuda.is_available() else "cpu"
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)

    torch.manual_seed(args.seed + rank)
    table_sizes = CRITEO_1TB_TABLE_SIZES
    if args.table_size_cap:
        table_sizes = [min(s, args.table_size_cap) for s in table_sizes]
    with torch.device(device):
        model = DLRM(table_sizes, embedding_dim=args.embedding_dim,
                     strategy=args.dist_strategy, dp_input=args.dp_input or world == 1)

    local_bs = args.batch_size // world
    feature_ids = model.local_cat_feature_ids()
    if args.dataset_path:
        data = RawBinaryDataset(args.dataset_path, batch_size=args.batch_size,
                                categorical_features=feature_ids,
                                categorical_feature_sizes=table_sizes,
                                rank=rank, world=world,
                                dp_input=args.dp_input or world == 1)
    else:
        data = SyntheticDLRMData(table_sizes, local_bs, num_batches=args.num_batches,
                                 device=device, rank=rank,
                                 feature_ids=feature_ids,
                                 dp_input=args.dp_input or world == 1,
                                 learnable=args.learnable_labels)

    # loss below is sum(BCE)/global_batch and dp grads are SUMMED
    # (average=False), so each grad is already the global-batch mean —
    # lr is used as-is (parity: reference lr=24 with a mean loss,
    # examples/dlrm/main.py).
    fused_modules = []
    if args.fused_optimizer:
        model.embeddings.enable_fused_sgd(args.learning_rate)
        fused_modules.append(model.embeddings)
    opt = de.DistributedOptimizer(SparseEmbeddingOptimizer(
        model.parameters(), lr=args.learning_rate, method="sgd"),
        average=False)
    sched = WarmupPolyDecay(opt, base_lr=args.learning_rate,
                            warmup_steps=args.warmup_steps,
                            decay_start=args.decay_start,
                            decay_steps=args.decay_steps,
                            fused_modules=fused_modules)
    de.broadcast_parameters(model)
    _loss_sum = torch.nn.BCEWithLogitsLoss(reduction="sum")
    loss_fn = lambda lg, lb: _loss_sum(lg, lb) / args.batch_size

    model.train()
    t0 = time.time()
    for step, (num, cats, labels) in enumerate(data):
        if step >= args.num_batches:
            break
        # set THIS step's lr before forward/backward: fused in-backward
        # updates consume the device-resident lr during backward, and the
        # dense optimizer reads it at opt.step() (warmup starts at step 0)
        sched.step()
        opt.zero_grad(set_to_none=True)
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=torch.cuda.is_available()):
            logits = model(num, cats)
            loss = loss_fn(logits.float(), labels)
        loss.backward()
        opt.step()
        if rank == 0 and step % 50 == 0:
            l = float(loss.detach())
            print(f"step {step:5d} loss {l:.4f} "
                  f"({args.batch_size * (step + 1) / (time.time() - t0):.0f} samples/s)")

    if args.eval:
        model.eval()
        scores, labels_all = [], []
        with torch.no_grad():
            for step, (num, cats, labels) in enumerate(data):
                if step >= 10:
                    break
                logits = model(num, cats)
                scores.append(torch.sigmoid(logits.float()).reshape(-1).cpu())
                labels_all.append(labels.reshape(-1).cpu())
        s = torch.cat(scores)
        l = torch.cat(labels_all)
        gathered_s = de.comm.all_gather_uneven(s)
        gathered_l = de.comm.all_gather_uneven(l)
        if rank == 0:
            print(f"AUC: {auc(torch.cat(gathered_s), torch.cat(gathered_l)):.5f}")

    if args.dump_embeddings:
        import numpy as np
        weights = model.embeddings.get_weights()
        if rank == 0:
            np.savez(args.dump_embeddings, *weights)
            print(f"embeddings dumped to {args.dump_embeddings}")
    if args.checkpoint_dir:
        de.save_embedding_checkpoint(model.embeddings, args.checkpoint_dir)
        if rank == 0:
            print(f"checkpoint written to {args.checkpoint_dir}")


if __name__ == "__main__":
    main()
