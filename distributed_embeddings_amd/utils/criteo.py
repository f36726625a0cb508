"""Criteo split-binary dataset reader + synthetic stand-in.

Capability parity with the reference ``examples/dlrm/utils.py:116-307``
(RawBinaryDataset): split-binary Criteo layout — one file per categorical
feature with per-feature integer dtype chosen by vocab size (int8/16/32), a
numerical fp16/fp32 file and a label int8 file — read with ``os.pread`` at
per-rank offsets, with model-parallel input mode reading only this rank's
features.  ``SyntheticDLRMData`` is the DummyDataset analog
(``utils.py:126-154``) generating random batches on device.
"""

import os
from typing import Optional, Sequence

import numpy as np
import torch


def _cat_dtype(vocab: int):
    # parity: reference utils.py:116-123 — smallest SIGNED dtype whose max
    # exceeds the vocab (int8 only below 127: ids are stored signed)
    if vocab < np.iinfo(np.int8).max:
        return np.int8, 1
    if vocab < np.iinfo(np.int16).max:
        return np.int16, 2
    return np.int32, 4


class RawBinaryDataset:
    """Iterates (numerical, cats, labels) local batches from split-binary files.

    Expected layout under ``data_path`` (train split):
      ``numerical.bin`` (fp16, [N, num_numerical]), ``label.bin`` (int8, [N]),
      ``cat_{i}.bin`` (per-feature dtype, [N]).
    """

    def __init__(self, data_path: str, batch_size: int,
                 categorical_features: Sequence[int],
                 categorical_feature_sizes: Sequence[int],
                 num_numerical: int = 13, rank: int = 0, world: int = 1,
                 dp_input: bool = True, valid: bool = False,
                 device=None, drop_last: bool = True):
        self.path = os.path.join(data_path, "validation" if valid else "train")
        self.batch_size = batch_size
        self.local_bs = batch_size // world
        self.rank, self.world = rank, world
        self.dp_input = dp_input
        self.num_numerical = num_numerical
        self.feature_ids = list(categorical_features)
        self.sizes = list(categorical_feature_sizes)
        self.device = device or (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))

        self._label_fd = os.open(os.path.join(self.path, "label.bin"), os.O_RDONLY)
        n_bytes = os.fstat(self._label_fd).st_size
        self.num_samples = n_bytes  # int8 labels
        self._num_fd = os.open(os.path.join(self.path, "numerical.bin"), os.O_RDONLY)
        self._cat_fds = {}
        for i in self.feature_ids:
            self._cat_fds[i] = os.open(os.path.join(self.path, f"cat_{i}.bin"),
                                       os.O_RDONLY)
        self.num_batches = self.num_samples // batch_size if drop_last else \
            (self.num_samples + batch_size - 1) // batch_size

    def __len__(self):
        return self.num_batches

    def _read(self, fd, offset_items, n_items, dtype, item_bytes, cols=1):
        raw = os.pread(fd, n_items * item_bytes * cols, offset_items * item_bytes * cols)
        arr = np.frombuffer(raw, dtype=dtype)
        if cols > 1:
            arr = arr.reshape(-1, cols)
        return arr

    def __iter__(self):
        for b in range(self.num_batches):
            base = b * self.batch_size
            if self.dp_input:
                start = base + self.rank * self.local_bs
                count = self.local_bs
            else:
                start = base       # model-parallel input: full global batch
                count = self.batch_size
            labels = self._read(self._label_fd, base + self.rank * self.local_bs,
                                self.local_bs, np.int8, 1)
            num = self._read(self._num_fd, base + self.rank * self.local_bs,
                             self.local_bs, np.float16, 2, cols=self.num_numerical)
            cats = []
            for i in self.feature_ids:
                dt, nb = _cat_dtype(self.sizes[i])
                ids = self._read(self._cat_fds[i], start, count, dt, nb)
                cats.append(torch.from_numpy(ids.astype(np.int64)).to(self.device))
            yield (torch.from_numpy(num.astype(np.float32)).to(self.device),
                   cats,
                   torch.from_numpy(labels.astype(np.float32)).to(self.device).unsqueeze(1))


class SyntheticDLRMData:
    """Random on-device batches shaped like Criteo (DummyDataset analog)."""

    def __init__(self, table_sizes: Sequence[int], local_bs: int,
                 num_batches: int = 100, num_numerical: int = 13,
                 device="cpu", rank: int = 0,
                 feature_ids: Optional[Sequence[int]] = None,
                 dp_input: bool = True, pool: int = 4, world: int = 1,
                 learnable: bool = False):
        """``learnable=True`` makes labels a deterministic function of the
        first categorical feature (id parity), so a correctly-scaled
        optimizer drives the BCE well below the ln(2) random floor — the
        convergence smoke for the lr contract (random labels cannot
        distinguish a working optimizer from a frozen one)."""
        from .input_gen import make_batch
        self.num_batches = num_batches
        self.pool = []
        feature_ids = list(feature_ids) if feature_ids is not None else \
            list(range(len(table_sizes)))
        bs = local_bs if dp_input else local_bs * max(world, 1)
        for i in range(pool):
            g = torch.Generator().manual_seed(17 + 131 * rank + i)
            sizes = [table_sizes[t] for t in feature_ids]
            cats = [c.to(device) for c in
                    make_batch(sizes, [1] * len(sizes), bs, generator=g)]
            num = torch.rand(local_bs, num_numerical, device=device)
            if learnable and cats:
                labels = (cats[0][:local_bs] % 2).view(-1, 1).float()
            else:
                labels = torch.randint(0, 2, (local_bs, 1), device=device).float()
            self.pool.append((num, cats, labels))

    def __len__(self):
        return self.num_batches

    def __iter__(self):
        for i in range(self.num_batches):
            yield self.pool[i % len(self.pool)]
