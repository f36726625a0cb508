#!/usr/bin/env python3
"""IntegerLookup end-to-end demo: raw int64/hex keys -> on-the-fly GPU vocab
-> embedding -> MLP, no offline preprocessing.

Capability parity with the reference ``examples/criteo/main.py:39-91``
(pandas TSV with hex-string categorical columns).  Without a TSV, random hex
keys are generated.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch import nn

import distributed_embeddings_amd as de


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--tsv", default=None, help="Criteo-format TSV (label, 13 int, 26 hex)")
    p.add_argument("--rows", type=int, default=100000, help="synthetic rows if no TSV")
    p.add_argument("--num-cat", type=int, default=26)
    p.add_argument("--max-tokens", type=int, default=100000)
    p.add_argument("--embedding-dim", type=int, default=32)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--batch-size", type=int, default=4096)
    return p.parse_args()


def load_data(args, device):
    if args.tsv:
        import pandas as pd
        df = pd.read_csv(args.tsv, sep="\t", header=None)
        labels = torch.tensor(df[0].values, dtype=torch.float32)
        cat_cols = []
        for c in range(14, 14 + args.num_cat):
            # hex string -> int64 (parity: reference main.py:39-41)
            keys = df[c].fillna("0").map(lambda s: int(str(s), 16) & ((1 << 63) - 1))
            cat_cols.append(torch.tensor(keys.values, dtype=torch.int64))
        cats = torch.stack(cat_cols, dim=1)
    else:
        g = torch.Generator().manual_seed(0)
        n = args.rows
        cats = torch.randint(0, 1 << 40, (n, args.num_cat), generator=g)
        labels = torch.randint(0, 2, (n,), generator=g).float()
    return cats.to(device), labels.to(device)


class Model(nn.Module):
    def __init__(self, num_cat, max_tokens, dim):
        super().__init__()
        self.lookups = nn.ModuleList(
            [de.IntegerLookup(max_tokens=max_tokens) for _ in range(num_cat)])
        self.embeddings = nn.ModuleList(
            [de.Embedding(max_tokens + 1, dim) for _ in range(num_cat)])
        self.mlp = nn.Sequential(nn.Linear(num_cat * dim, 128), nn.ReLU(),
                                 nn.Linear(128, 1))

    def forward(self, cats):
        outs = []
        for i, (lk, emb) in enumerate(zip(self.lookups, self.embeddings)):
            outs.append(emb(lk(cats[:, i])))
        return self.mlp(torch.cat(outs, dim=1))


def main():
    args = parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    cats, labels = load_data(args, device)
    model = Model(args.num_cat, args.max_tokens, args.embedding_dim).to(device)
    opt = torch.optim.Adagrad(model.parameters(), lr=0.05)
    loss_fn = nn.BCEWithLogitsLoss()
    n = cats.shape[0]
    args.batch_size = min(args.batch_size, n)
    for epoch in range(args.epochs):
        for s in range(0, n - args.batch_size + 1, args.batch_size):
            batch = cats[s:s + args.batch_size]
            lb = labels[s:s + args.batch_size].unsqueeze(1)
            opt.zero_grad()
            loss = loss_fn(model(batch), lb)
            loss.backward()
            opt.step()
        print(f"epoch {epoch}: loss {float(loss.detach()):.4f}, "
              f"vocab[0] size {model.lookups[0].vocabulary_size()}")


if __name__ == "__main__":
    main()
