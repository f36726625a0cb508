"""Optimizers with fused O(nnz) sparse embedding updates.

The lookup backward emits coalesced ``(unique_ids, unique_grad)`` sparse COO
gradients (the IndexedSlices contract).  Stock torch optimizers re-coalesce
every sparse grad (a device-wide merge sort + segmented add per step); the
fused path here applies the rows directly with one HIP kernel
(``sparse_row_update`` in ``csrc/embedding_ops.hip``).

``SparseEmbeddingOptimizer`` handles both sparse-grad embedding tables and
dense-grad params (MLP weights) in one optimizer; it is the MI355X analog of
the reference examples' per-variable optimizer usage (SGD in
``examples/dlrm/utils.py:45-88``, Adagrad in
``examples/benchmarks/synthetic_models/main.py``).
"""

import torch

from ..ops import _backend


class SparseEmbeddingOptimizer(torch.optim.Optimizer):
    """SGD / Adagrad with fused sparse row updates.

    Args:
      params: iterable of parameters (dense and sparse-grad mixed).
      lr: learning rate.
      method: ``"sgd"`` or ``"adagrad"``.
      eps: adagrad epsilon.
      assume_coalesced: trust that every sparse grad holds unique, sorted ids
        (true when each embedding parameter receives exactly one lookup
        backward per step — the fused-group design guarantees it).  Set False
        if a table is looked up multiple times per step outside the fused
        path.
    """

    def __init__(self, params, lr: float = 0.01, method: str = "sgd",
                 eps: float = 1e-10, assume_coalesced: bool = True):
        if method not in ("sgd", "adagrad"):
            raise ValueError(f"unknown method {method!r}")
        defaults = dict(lr=lr, method=method, eps=eps,
                        assume_coalesced=assume_coalesced)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            eps = group["eps"]
            adagrad = group["method"] == "adagrad"
            dense = []
            for p in group["params"]:
                g = p.grad
                if g is None:
                    continue
                if g.layout == torch.sparse_coo:
                    self._sparse_update(p, g, lr, eps, adagrad,
                                        group["assume_coalesced"])
                else:
                    dense.append(p)
            if dense:
                self._dense_update_batch(dense, lr, eps, adagrad)
        return loss

    def _state_for(self, p):
        st = self.state[p]
        if "sum" not in st:
            # fp32 accumulator state regardless of param storage dtype
            st["sum"] = torch.zeros(p.shape, dtype=torch.float32,
                                    device=p.device)
        return st["sum"]

    def _sparse_update(self, p, g, lr, eps, adagrad, assume_coalesced):
        if not (assume_coalesced or g.is_coalesced()):
            g = g.coalesce()
        ids = g._indices()[0]
        vals = g._values().float()
        if p.is_cuda:
            state = self._state_for(p) if adagrad else torch.empty(0)
            _backend.ops().sparse_row_update(p.data, state, ids.contiguous(),
                                             vals.contiguous(), lr, eps,
                                             adagrad)
        else:
            if adagrad:
                state = self._state_for(p)
                state.index_add_(0, ids, vals * vals)
                denom = state.index_select(0, ids).sqrt_().add_(eps)
                p.data.index_add_(0, ids, (-lr * vals / denom).to(p.dtype))
            else:
                p.data.index_add_(0, ids, (-lr * vals).to(p.dtype))

    def _dense_update_batch(self, params, lr, eps, adagrad):
        """foreach-batched dense updates (one fused launch set per step)."""
        grads = [p.grad for p in params]
        if adagrad:
            states = [self._state_for(p) for p in params]
            torch._foreach_addcmul_(states, grads, grads, value=1.0)
            denoms = torch._foreach_sqrt(states)
            torch._foreach_add_(denoms, eps)
            torch._foreach_addcdiv_([p.data for p in params], grads, denoms,
                                    value=-lr)
        else:
            torch._foreach_add_([p.data for p in params], grads, alpha=-lr)
