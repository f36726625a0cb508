"""Checkpoint helpers for DistributedEmbedding.

The checkpoint CONTRACT (reference parity) is "global per-table fp32 numpy
arrays in original table order" — ``get_weights``/``set_weights`` implement
it; these helpers add the directory layout: one ``table_{i:05d}.npy`` per
table, written by rank 0, loaded via the ``set_weights`` path branch (which
``np.load``-mmaps each file and copies only this rank's slices, in
``chunk_elements``-bounded pieces).
"""

import os
from typing import List

import numpy as np

from . import comm


def save_embedding_checkpoint(dist_embedding, directory: str,
                              chunk_elements: int = 128 * 1024 * 1024) -> List[str]:
    """Reassembles full tables (collective — call on EVERY rank) and writes
    one ``.npy`` per table from rank 0.  Returns the file paths."""
    weights = dist_embedding.get_weights(all_ranks=False,
                                         chunk_elements=chunk_elements)
    paths = [os.path.join(directory, f"table_{t:05d}.npy")
             for t in range(len(weights))]
    if comm.rank() == 0:
        os.makedirs(directory, exist_ok=True)
        for p, w in zip(paths, weights):
            np.save(p, w)
    comm.barrier()
    return paths


def load_embedding_checkpoint(dist_embedding, directory: str,
                              chunk_elements: int = 128 * 1024 * 1024) -> None:
    """Loads a directory written by :func:`save_embedding_checkpoint`.

    Every rank mmap-reads only its own slices (no broadcast, no full-table
    materialization — parity with the reference's ``np.load(mmap_mode='r')``
    path, dist_model_parallel.py:911-950)."""
    names = sorted(f for f in os.listdir(directory)
                   if f.startswith("table_") and f.endswith(".npy"))
    n_tables = len(dist_embedding.strategy.configs)
    if len(names) != n_tables:
        raise ValueError(
            f"checkpoint has {len(names)} tables, model expects {n_tables}")
    dist_embedding.set_weights(
        [os.path.join(directory, f) for f in names],
        chunk_elements=chunk_elements)
