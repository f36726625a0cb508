"""IntegerLookup — on-the-fly vocabulary build + lookup for raw int64 keys.

MI355X-native equivalent of the reference layer
(``/root/reference/distributed_embeddings/python/layers/embedding.py:202-281``).
On GPU the vocabulary lives in a hand-written open-addressing (linear-probe)
int64 hash resident in module buffers — replacing the reference's
cuCollections ``static_map`` (``embedding_lookup_kernels.cu:383-516``) with a
two-kernel CDNA4 design (free-slot scan, then insert-and-find with
device-scope 64-bit atomicCAS).  On CPU a dict-based path with identical
semantics is used (parity: the reference's ``DenseHashTable`` CPU path,
``embedding.py:228-253``).

Semantics:
* value 0 is reserved for OOV / overflow (slot pre-claimed at init,
  parity ``embedding.py:217-220``);
* new keys are assigned the next free value (1, 2, ...) on first sight;
* when the table is full, unseen keys map to 0;
* per-value frequency counts are maintained (``counts`` buffer).
"""

from typing import List

import torch
from torch import nn

from ..ops import _backend

_LOAD_FACTOR = 1.5  # capacity multiplier, parity: reference embedding.py:226


class IntegerLookup(nn.Module):
    """Maps arbitrary int64 keys to a dense [0, max_tokens] vocabulary."""

    def __init__(self, max_tokens: int = 100000, device=None):
        super().__init__()
        if max_tokens <= 0:
            raise ValueError("max_tokens must be positive")
        self.max_tokens = int(max_tokens)
        self.capacity = int(_LOAD_FACTOR * (self.max_tokens + 1))
        # GPU-resident hash state (also the single source of truth for
        # checkpointing; CPU path mirrors into it lazily).
        self.register_buffer("table_keys", torch.full((self.capacity,), -1, dtype=torch.int64,
                                                      device=device))
        self.register_buffer("table_values", torch.zeros(self.capacity, dtype=torch.int64,
                                                         device=device))
        # counts[v] = frequency of value v; counts[0] pre-claimed for OOV.
        counts = torch.zeros(self.max_tokens + 1, dtype=torch.int32, device=device)
        counts[0] = 1
        self.register_buffer("counts", counts)
        self._cpu_map = None  # lazy dict for the CPU path

    # ------------------------------------------------------------------ CPU

    def _cpu_state(self):
        if self._cpu_map is None:
            self._cpu_map = {}
            keys = self.table_keys.cpu()
            vals = self.table_values.cpu()
            occupied = keys >= 0
            for k, v in zip(keys[occupied].tolist(), vals[occupied].tolist()):
                self._cpu_map[k] = v
        return self._cpu_map

    def _forward_cpu(self, keys: torch.Tensor) -> torch.Tensor:
        m = self._cpu_state()
        counts = self.counts
        next_val = int((counts > 0).sum().item())
        out = torch.empty_like(keys)
        flat_in = keys.reshape(-1)
        flat_out = out.reshape(-1)
        for i, k in enumerate(flat_in.tolist()):
            v = m.get(k)
            if v is None:
                if next_val <= self.max_tokens:
                    v = next_val
                    m[k] = v
                    next_val += 1
                    # persist into the hash buffers (linear probe)
                    self._cpu_insert(k, v)
                else:
                    v = 0
            counts[v] += 1
            flat_out[i] = v
        return out

    @staticmethod
    def _mix64(k: int) -> int:
        """splitmix64 finalizer — must match ``mix64`` in embedding_ops.hip so
        CPU-inserted state probes identically on GPU."""
        m = (1 << 64) - 1
        k = (k + 0x9E3779B97F4A7C15) & m
        k = ((k ^ (k >> 30)) * 0xBF58476D1CE4E5B9) & m
        k = ((k ^ (k >> 27)) * 0x94D049BB133111EB) & m
        return (k ^ (k >> 31)) & m

    def _cpu_insert(self, key: int, value: int):
        cap = self.capacity
        slot = self._mix64(key & ((1 << 64) - 1)) % cap
        while int(self.table_keys[slot]) != -1:
            slot = (slot + 1) % cap
        self.table_keys[slot] = key
        self.table_values[slot] = value

    # ------------------------------------------------------------------ GPU

    def forward(self, keys: torch.Tensor) -> torch.Tensor:
        if keys.dtype != torch.int64:
            keys = keys.long()
        if keys.is_cuda:
            return _backend.ops().integer_lookup(
                keys.contiguous().reshape(-1), self.table_keys, self.table_values,
                self.counts, self.max_tokens).view(keys.shape)
        return self._forward_cpu(keys)

    # ------------------------------------------------------------- inspection

    def vocabulary_size(self) -> int:
        return int((self.counts > 0).sum().item())

    def get_vocabulary(self) -> List[int]:
        """Keys in value order; value 0 (OOV) reported as -1 placeholder.

        Parity: reference ``get_vocabulary`` (``embedding.py:255-281``).
        """
        keys = self.table_keys.cpu()
        vals = self.table_values.cpu()
        occupied = keys >= 0
        pairs = sorted(zip(vals[occupied].tolist(), keys[occupied].tolist()))
        vocab = [-1]
        vocab.extend(k for _, k in pairs)
        return vocab
