"""Embedding layers (single-GPU building blocks).

MI355X-native equivalents of the reference Keras layers in
``/root/reference/distributed_embeddings/python/layers/embedding.py``:

* :class:`Embedding`  — unified dense/ragged/sparse lookup + combiner
  (parity: ``embedding.py:50-170``).
* :class:`ConcatOneHotEmbedding` — all-one-hot tables fused into one variable
  with cumulative row offsets (parity: ``embedding.py:173-198``).
"""

import math
from typing import Callable, Optional, Union

import torch
from torch import nn

from ..ops.embedding_lookup import (Ragged, csr_lookup_fused_optimizer,
                                    embedding_lookup)


def _default_init(weight: torch.Tensor) -> None:
    # Matches the reference default keras 'uniform' initializer range.
    nn.init.uniform_(weight, -0.05, 0.05)


def scaled_uniform_init(weight: torch.Tensor) -> None:
    """DLRM-style uniform(-1/sqrt(rows), 1/sqrt(rows)) initializer.

    Parity: reference ``examples/dlrm/utils.py:27-41`` (DLRMInitializer).
    """
    bound = 1.0 / math.sqrt(weight.shape[0])
    nn.init.uniform_(weight, -bound, bound)


class Embedding(nn.Module):
    """An embedding lookup layer with optional row combiner.

    Args:
      input_dim: vocabulary size (number of rows).
      output_dim: embedding width (number of columns).
      combiner: ``None``, ``'sum'`` or ``'mean'``.  With a combiner, the
        hotness (last) dimension of the input is reduced away.
      initializer: callable applied to the weight at construction.
      dtype: parameter dtype (default fp32).

    Input/output shapes (parity: reference ``embedding.py:65-69``):
      * dense ``[d0, ..., dn]`` ids, no combiner -> ``[d0, ..., dn, output_dim]``
      * dense ``[d0, ..., dn, hotness]`` + combiner -> ``[d0, ..., dn, output_dim]``
      * ``Ragged`` (2-D CSR) + combiner -> ``[nrows, output_dim]``
      * ``torch.sparse_coo`` 2-D ids + combiner -> ``[nrows, output_dim]``
    """

    def __init__(
        self,
        input_dim: int,
        output_dim: int,
        combiner: Optional[str] = None,
        initializer: Optional[Callable[[torch.Tensor], None]] = None,
        dtype: torch.dtype = torch.float32,
        device=None,
        sparse_grad: bool = True,
    ):
        super().__init__()
        if input_dim <= 0 or output_dim <= 0:
            raise ValueError("input_dim and output_dim must be positive")
        if combiner not in (None, "sum", "mean"):
            raise ValueError(f"invalid combiner {combiner!r}")
        self.input_dim = int(input_dim)
        self.output_dim = int(output_dim)
        self.combiner = combiner
        # sparse_grad: plain-gather lookups route through the CSR kernel so the
        # weight grad is a coalesced sparse tensor (IndexedSlices contract) —
        # a dense grad for a 288 GB-class table is not an option.
        self.sparse_grad = sparse_grad
        self.weight = nn.Parameter(torch.empty(input_dim, output_dim, dtype=dtype, device=device))
        (initializer or _default_init)(self.weight)
        # Internal: row-slice shards tolerate out-of-range ids (contribute a
        # zero row) so only the owning shard is in-bounds — parity with the
        # reference's reliance on TF GPU OOB-gather-zeros
        # (dist_model_parallel.py:889-904).
        self._oob_zero = False

    def extra_repr(self) -> str:
        return f"input_dim={self.input_dim}, output_dim={self.output_dim}, combiner={self.combiner}"

    def _apply(self, fn, recurse=True):
        # CPU-offloaded tables are pinned: .to('cuda') / .cuda() on the parent
        # module must not move the weight (parity: reference builds offloaded
        # variables under tf.device('CPU:0'), dist_model_parallel.py:1186-1189).
        if getattr(self, "_cpu_offload", False):
            return self
        return super()._apply(fn, recurse)

    def enable_fused_sgd(self, lr: float):
        """In-backward SGD (see enable_fused_optimizer)."""
        return self.enable_fused_optimizer("sgd", lr)

    def enable_fused_optimizer(self, method: str, lr: float, eps: float = 1e-10):
        """In-backward fused optimizer: the lookup's backward applies the
        SGD/Adagrad update directly (no grad tensor, no host sync; the step
        becomes hipGraph-capturable).  The training optimizer must not also
        update this weight (its ``.grad`` stays ``None``)."""
        if method not in ("sgd", "adagrad"):
            raise ValueError(f"unknown fused optimizer {method!r}")
        for name in ("_fused_lr", "_fused_state"):  # allow re-configuring
            if name in self._buffers:
                del self._buffers[name]
        self.register_buffer("_fused_lr",
                             torch.tensor([float(lr)], dtype=torch.float32,
                                          device=self.weight.device))
        self._fused_method = method
        self._fused_eps = float(eps)
        if method == "adagrad":
            # fp32 state regardless of table storage dtype
            self.register_buffer("_fused_state",
                                 torch.zeros(self.weight.shape,
                                             dtype=torch.float32,
                                             device=self.weight.device))
        else:
            self.register_buffer("_fused_state",
                                 torch.empty(0, dtype=torch.float32,
                                             device=self.weight.device))
        # optimizer state is shard-local: never broadcast/allreduce it (the
        # weight itself carries de_local when model-parallel)
        self._fused_state.de_local = getattr(self.weight, "de_local", False)
        self._fused_lr.de_local = False  # lr is global; broadcasting is fine
        return self

    def set_fused_lr(self, lr: float):
        if getattr(self, "_fused_lr", None) is None:
            raise RuntimeError("enable_fused_sgd() first")
        with torch.no_grad():
            self._fused_lr.fill_(float(lr))

    def csr_lookup(self, values: torch.Tensor, row_splits: torch.Tensor,
                   combiner: str, out_dtype=None) -> torch.Tensor:
        """CSR lookup through this layer (fused-SGD aware).

        ``out_dtype=torch.bfloat16`` makes the HIP kernel store bf16 directly
        (fp32 accumulation; backward consumes bf16 grads natively) — no
        separate cast kernel on the hot path."""
        if getattr(self, "_fused_lr", None) is not None and self.training:
            return csr_lookup_fused_optimizer(
                self.weight, values, row_splits, combiner, self._fused_lr,
                self._fused_state, self._fused_method == "adagrad",
                self._fused_eps, out_dtype)
        from ..ops.embedding_lookup import _CsrLookup
        return _CsrLookup.apply(self.weight, values, row_splits, combiner,
                                out_dtype)

    def get_config(self) -> dict:
        """Planner-facing config (reference uses keras ``get_config()``)."""
        return {
            "input_dim": self.input_dim,
            "output_dim": self.output_dim,
            "combiner": self.combiner,
        }

    def _gather(self, ids: torch.Tensor) -> torch.Tensor:
        flat = ids.reshape(-1)
        if self.sparse_grad and (self.weight.requires_grad or self._oob_zero):
            # hotness-1 CSR: same gather, sparse (IndexedSlices-style) grad,
            # OOB ids contribute zero rows.
            splits = torch.arange(flat.numel() + 1, device=flat.device, dtype=torch.long)
            out = self.csr_lookup(flat, splits, "sum")
        elif self._oob_zero:
            valid = (flat >= 0) & (flat < self.input_dim)
            safe = torch.where(valid, flat, torch.zeros_like(flat))
            out = self.weight.index_select(0, safe)
            out = out * valid.unsqueeze(1).to(out.dtype)
        else:
            out = self.weight.index_select(0, flat)
        return out.view(*ids.shape, self.output_dim)

    def forward(self, ids: Union[torch.Tensor, Ragged]) -> torch.Tensor:
        if isinstance(ids, Ragged):
            if self.combiner is None:
                raise ValueError("Ragged input requires a combiner")
            return self.csr_lookup(ids.values.long(), ids.row_splits.long(),
                                    self.combiner)

        if ids.layout == torch.sparse_coo:
            if self.combiner is None:
                raise ValueError("Sparse input requires a combiner")
            return embedding_lookup(self.weight, ids, self.combiner)

        if ids.dtype != torch.long:
            # parity: reference casts non-int inputs (embedding.py:121-123)
            ids = ids.long()

        if self.combiner is None:
            return self._gather(ids)

        if ids.dim() < 2:
            raise ValueError("ids with a combiner must have a hotness dimension "
                             "(parity: reference embedding.py:133-135)")
        lead_shape = ids.shape[:-1]
        flat2d = ids.reshape(-1, ids.shape[-1])
        if getattr(self, "_fused_lr", None) is not None and self.training:
            # fused in-backward optimizer: dense [b, h] as CSR so the update
            # applies here too (not only for Ragged inputs)
            b, h = flat2d.shape
            splits = torch.arange(b + 1, device=flat2d.device,
                                  dtype=torch.long) * h
            out = self.csr_lookup(flat2d.reshape(-1), splits, self.combiner)
        elif self._oob_zero or flat2d.shape[1] > 1:
            out = embedding_lookup(self.weight, flat2d, self.combiner)
        else:
            out = self._gather(flat2d.reshape(-1))
        return out.view(*lead_shape, self.output_dim)


class ConcatOneHotEmbedding(nn.Module):
    """Multiple one-hot (hotness-1) tables fused into a single variable.

    Inputs of shape ``[batch, num_tables]`` are offset by per-table cumulative
    row starts and looked up with a single gather.
    Parity: reference ``embedding.py:173-198``.
    """

    def __init__(self, table_sizes, output_dim, initializer=None, dtype=torch.float32,
                 device=None):
        super().__init__()
        self.table_sizes = [int(s) for s in table_sizes]
        self.output_dim = int(output_dim)
        total = sum(self.table_sizes)
        self.weight = nn.Parameter(torch.empty(total, output_dim, dtype=dtype, device=device))
        (initializer or _default_init)(self.weight)
        offsets = torch.cumsum(torch.tensor([0] + self.table_sizes[:-1], dtype=torch.long), 0)
        self.register_buffer("offsets", offsets)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if ids.dim() != 2 or ids.shape[1] != len(self.table_sizes):
            raise ValueError(
                f"expected ids [batch, {len(self.table_sizes)}], got {tuple(ids.shape)}")
        shifted = ids.long() + self.offsets.unsqueeze(0)
        return self.weight.index_select(0, shifted.reshape(-1)).view(
            ids.shape[0], len(self.table_sizes), self.output_dim)
