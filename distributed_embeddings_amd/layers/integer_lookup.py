"""IntegerLookup — on-the-fly vocabulary build + lookup for raw int64 keys.

MI355X-native equivalent of the reference layer
(``/root/reference/distributed_embeddings/python/layers/embedding.py:202-281``).
On GPU the vocabulary lives in a hand-written open-addressing (linear-probe)
int64 hash resident in module buffers — replacing the reference's
cuCollections ``static_map`` (``embedding_lookup_kernels.cu:383-516``) with a
two-kernel CDNA4 design (free-slot scan, then insert-and-find with
device-scope 64-bit atomicCAS).  The CPU path is fully vectorized (numpy
probing over the same buffer layout — parity with the reference's
``DenseHashTable`` CPU path, ``embedding.py:228-253``, which is tensorized,
not a Python loop).

Semantics:
* value 0 is reserved for OOV / overflow (slot pre-claimed at init,
  parity ``embedding.py:217-220``);
* new keys are assigned the next free value (1, 2, ...) on first sight;
* when the table is full, unseen keys map to 0 (with ``auto_grow=True`` the
  table instead rehashes into doubled capacity — an MI355X extension beyond
  the reference's fixed ``max_tokens`` guess);
* per-value frequency counts are maintained (``counts`` buffer).
"""

from typing import List

import numpy as np
import torch
from torch import nn

from ..ops import _backend

_LOAD_FACTOR = 1.5  # capacity multiplier, parity: reference embedding.py:226
_GROW_AT = 0.8      # auto_grow trigger: assigned values / max_tokens


def _mix64_np(k: np.ndarray) -> np.ndarray:
    """splitmix64 finalizer on uint64 arrays — must match ``mix64`` in
    embedding_ops.hip so CPU-inserted state probes identically on GPU."""
    k = k.astype(np.uint64, copy=True)
    k += np.uint64(0x9E3779B97F4A7C15)
    k = (k ^ (k >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    k = (k ^ (k >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return k ^ (k >> np.uint64(31))


class IntegerLookup(nn.Module):
    """Maps arbitrary int64 keys to a dense [0, max_tokens] vocabulary.

    Args:
      max_tokens: vocabulary budget (value 0 is reserved for OOV).
      auto_grow: if True, reaching 80% load (or overflowing) doubles
        ``max_tokens`` and rehashes instead of mapping new keys to OOV.
        Growth checks track a sound upper bound on the vocabulary size, so
        the device sync happens only when that bound nears capacity
        (amortized away by each doubling); False = fixed capacity
        (reference parity).
    """

    def __init__(self, max_tokens: int = 100000, auto_grow: bool = False,
                 device=None):
        super().__init__()
        if max_tokens <= 0:
            raise ValueError("max_tokens must be positive")
        self.max_tokens = int(max_tokens)
        self.auto_grow = bool(auto_grow)
        self.capacity = int(_LOAD_FACTOR * (self.max_tokens + 1))
        # GPU-resident hash state (also the single source of truth for
        # checkpointing; the CPU path probes the same layout via numpy views).
        self.register_buffer("table_keys", torch.full((self.capacity,), -1, dtype=torch.int64,
                                                      device=device))
        self.register_buffer("table_values", torch.zeros(self.capacity, dtype=torch.int64,
                                                         device=device))
        # counts[v] = frequency of value v; counts[0] pre-claimed for OOV.
        counts = torch.zeros(self.max_tokens + 1, dtype=torch.int32, device=device)
        counts[0] = 1
        self.register_buffer("counts", counts)
        # auto_grow bookkeeping: vocabulary_size() <= _known_used + _unseen
        # always holds (each looked-up element can add at most one value), so
        # growth checks only SYNC when this upper bound nears capacity —
        # steady state runs without any device sync.
        self._known_used = 1
        self._unseen = 0

    # ------------------------------------------------------------------ CPU

    def _probe_np(self, uniq: np.ndarray):
        """Vectorized linear probe: per unique key returns
        (value, found, slot-of-hit)."""
        tk = self.table_keys.numpy()
        tv = self.table_values.numpy()
        cap = self.capacity
        slot = (_mix64_np(uniq) % np.uint64(cap)).astype(np.int64)
        vals = np.zeros(len(uniq), dtype=np.int64)
        found = np.zeros(len(uniq), dtype=bool)
        hit_slot = np.full(len(uniq), -1, dtype=np.int64)
        active = np.arange(len(uniq))
        for _ in range(cap):
            if not active.size:
                break
            s = slot[active]
            k = tk[s]
            hit = k == uniq[active]
            empty = k == -1
            if hit.any():
                vals[active[hit]] = tv[s[hit]]
                found[active[hit]] = True
                hit_slot[active[hit]] = s[hit]
            cont = ~(hit | empty)
            active = active[cont]
            slot[active] = (slot[active] + 1) % cap
        return vals, found, hit_slot

    def _insert_np(self, keys_np: np.ndarray, vals_np: np.ndarray):
        """Vectorized insert of UNIQUE keys with pre-assigned values.

        Collision rounds: contenders probe, first contender per free slot
        wins, losers (and occupied-slot probes) advance — each round is pure
        numpy over the still-pending set.
        """
        tk = self.table_keys.numpy()
        tv = self.table_values.numpy()
        cap = self.capacity
        slot = (_mix64_np(keys_np) % np.uint64(cap)).astype(np.int64)
        pending = np.arange(len(keys_np))
        while pending.size:
            s = slot[pending]
            occupied = tk[s] != -1
            cand = pending[~occupied]
            if cand.size:
                cs = s[~occupied]
                order = np.argsort(cs, kind="stable")
                cs_o, cand_o = cs[order], cand[order]
                first = np.ones(cs_o.size, dtype=bool)
                first[1:] = cs_o[1:] != cs_o[:-1]
                win, ws = cand_o[first], cs_o[first]
                tk[ws] = keys_np[win]
                tv[ws] = vals_np[win]
                lose = cand_o[~first]
            else:
                lose = pending[:0]
            adv = np.concatenate([pending[occupied], lose])
            slot[adv] = (slot[adv] + 1) % cap
            pending = adv

    def _forward_cpu(self, keys: torch.Tensor) -> torch.Tensor:
        flat = keys.reshape(-1).numpy()
        uniq, first_idx, inverse = np.unique(flat, return_index=True,
                                             return_inverse=True)
        vals, found, hit_slot = self._probe_np(uniq)
        miss = np.flatnonzero(~found)
        # under auto_grow, keys a full GPU table once pinned to value 0 are
        # assignable again (their slot value upgrades in place)
        upgrade = np.flatnonzero(found & (vals == 0)) if self.auto_grow \
            else np.empty(0, dtype=np.int64)
        need = np.concatenate([miss, upgrade])
        if need.size:
            # assign values in first-occurrence order (reference CPU parity)
            need = need[np.argsort(first_idx[need], kind="stable")]
            next_val = self.vocabulary_size()
            if self.auto_grow:
                while next_val + need.size > self.max_tokens + 1 or \
                        next_val + need.size > _GROW_AT * self.max_tokens:
                    self._grow()
            n_assign = max(0, min(need.size, self.max_tokens + 1 - next_val))
            take = need[:n_assign]
            new_vals = next_val + np.arange(n_assign, dtype=np.int64)
            vals[take] = new_vals
            ins = take[hit_slot[take] < 0]
            upg = take[hit_slot[take] >= 0]
            if ins.size:
                self._insert_np(uniq[ins], vals[ins])
            if upg.size:
                self.table_values.numpy()[hit_slot[upg]] = vals[upg]
            # overflow (auto_grow off): remaining keys stay at value 0
        out = vals[inverse]
        self._unseen += int(flat.size)
        self.counts += torch.from_numpy(
            np.bincount(out, minlength=self.counts.numel()).astype(np.int32))
        return torch.from_numpy(out).view(keys.shape)

    # ------------------------------------------------------------------ grow

    def _grow(self):
        """Double max_tokens, rehash all pairs into the larger table."""
        self.max_tokens *= 2
        new_cap = int(_LOAD_FACTOR * (self.max_tokens + 1))
        dev = self.table_keys.device
        new_keys = torch.full((new_cap,), -1, dtype=torch.int64, device=dev)
        new_vals = torch.zeros(new_cap, dtype=torch.int64, device=dev)
        if self.table_keys.is_cuda:
            _backend.ops().hash_rehash(self.table_keys, self.table_values,
                                       new_keys, new_vals)
        else:
            tk = self.table_keys.numpy()
            # drop value-0 (overflow-pinned) entries — parity with the GPU
            # hash_reinsert kernel
            occ = (tk != -1) & (self.table_values.numpy() != 0)
            old_k = tk[occ].copy()
            old_v = self.table_values.numpy()[occ].copy()
            self.table_keys = new_keys
            self.table_values = new_vals
            self.capacity = new_cap
            self._insert_np(old_k, old_v)
            # counts extension below; keys/values already swapped
            new_counts = torch.zeros(self.max_tokens + 1, dtype=torch.int32)
            new_counts[:self.counts.numel()] = self.counts
            self.counts = new_counts
            return
        self.table_keys = new_keys
        self.table_values = new_vals
        self.capacity = new_cap
        new_counts = torch.zeros(self.max_tokens + 1, dtype=torch.int32,
                                 device=dev)
        new_counts[:self.counts.numel()] = self.counts
        self.counts = new_counts

    # ------------------------------------------------------------------ GPU

    def forward(self, keys: torch.Tensor) -> torch.Tensor:
        if keys.dtype != torch.int64:
            keys = keys.long()
        if not keys.is_cuda:
            return self._forward_cpu(keys)
        n_keys = keys.numel()
        if self.auto_grow:
            # proactive growth at 80% load; the sync happens only when the
            # sound upper bound (known_used + elements seen since the last
            # exact count) approaches it — amortized away by each doubling
            if self._known_used + self._unseen + n_keys > \
                    _GROW_AT * self.max_tokens:
                used = self.vocabulary_size()  # one D2H sync (rare)
                self._known_used, self._unseen = used, 0
                while used > _GROW_AT * self.max_tokens:
                    self._grow()
        out = _backend.ops().integer_lookup(
            keys.contiguous().reshape(-1), self.table_keys, self.table_values,
            self.counts, self.max_tokens)
        self._unseen += n_keys
        if self.auto_grow and \
                self._known_used + self._unseen >= self.max_tokens + 1:
            used = self.vocabulary_size()
            self._known_used, self._unseen = used, 0
            if used >= self.max_tokens + 1:
                # Table filled mid-batch: out==0 elements are unresolved keys
                # (in auto_grow mode nothing is true-OOV).  Retract only
                # their counts[0] contribution, grow, resolve just that
                # subset — resolved keys keep their counts, so the counts>0
                # free-value invariant holds throughout.
                unresolved = out == 0
                n0 = int(unresolved.sum().item())
                if n0:
                    # counts[0] retraction does NOT change vocabulary_size()
                    # (value 0 stays pre-claimed), so _known_used is kept
                    self.counts[0] -= n0
                    self._grow()
                    out = out.clone()
                    out[unresolved] = self.forward(
                        keys.contiguous().reshape(-1)[unresolved]).reshape(-1)
        return out.view(keys.shape)

    # --------------------------------------------------------- (de)serialize

    def _load_from_state_dict(self, state_dict, prefix, *args, **kwargs):
        # a grown checkpoint may be larger than this module's fresh buffers:
        # resize to the incoming shapes before the standard copy
        tk = state_dict.get(prefix + "table_keys")
        if tk is not None and tk.numel() != self.table_keys.numel():
            dev = self.table_keys.device
            self.capacity = tk.numel()
            self.table_keys = torch.empty_like(tk, device=dev)
            self.table_values = torch.empty_like(tk, device=dev)
        cn = state_dict.get(prefix + "counts")
        if cn is not None and cn.numel() != self.counts.numel():
            self.max_tokens = cn.numel() - 1
            self.counts = torch.empty_like(cn, device=self.counts.device)
        super()._load_from_state_dict(state_dict, prefix, *args, **kwargs)

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
