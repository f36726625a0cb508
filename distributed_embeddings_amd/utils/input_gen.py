"""Synthetic categorical input generation.

Parity: reference ``InputGenerator`` power-law ids + pre-generated batch pool
(``examples/benchmarks/synthetic_models/synthetic_models.py:31-113``).
"""

from typing import List, Optional, Sequence

import torch


def power_law_ids(vocab: int, shape, alpha: float = 1.05,
                  generator: Optional[torch.Generator] = None,
                  device=None) -> torch.Tensor:
    """Draws ids with a power-law CDF (id^(1-alpha) tail), id in [0, vocab)."""
    u = torch.rand(shape, generator=generator, device=device)
    if alpha == 1.0:
        ids = torch.exp(u * torch.log(torch.tensor(float(vocab)))) - 1.0
    else:
        # inverse-CDF of p(x) ~ x^-alpha on [1, vocab]
        one_m = 1.0 - alpha
        hi = float(vocab) ** one_m
        ids = (u * (hi - 1.0) + 1.0) ** (1.0 / one_m) - 1.0
    return ids.clamp_(0, vocab - 1).long()


def uniform_ids(vocab: int, shape, generator=None, device=None) -> torch.Tensor:
    return torch.randint(0, vocab, shape, generator=generator, device=device)


def make_batch(table_sizes: Sequence[int], hotness: Sequence[int], batch: int,
               alpha: float = 1.05, device=None,
               generator: Optional[torch.Generator] = None,
               keep_hot_dim: bool = False) -> List[torch.Tensor]:
    """One batch of categorical inputs: [batch] for hotness-1 else [batch, h].

    ``keep_hot_dim=True`` emits [batch, 1] for hotness-1 (combiner layers
    require a hotness dimension, parity: reference embedding.py:133-135).
    """
    out = []
    for size, h in zip(table_sizes, hotness):
        shape = (batch,) if h == 1 and not keep_hot_dim else (batch, h)
        if alpha and size > 1:
            out.append(power_law_ids(size, shape, alpha, generator, device))
        else:
            out.append(uniform_ids(size, shape, generator, device))
    return out
