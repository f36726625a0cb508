"""Fused Linear+ReLU for the DLRM MLPs.

On GPU the forward runs as ONE hipBLASLt call with a bias+ReLU epilogue
(``torch._addmm_activation`` — measured free vs plain addmm on MI355X, see
``tools/probe_addmm_act.py``), removing the separate ReLU kernel from every
MLP layer.  The backward recovers the ReLU mask from the saved output
(``out > 0``), so no pre-activation tensor is kept.

CPU falls back to ``relu(linear(x))`` with identical semantics.
"""

import math

import torch
from torch import nn

try:  # torch >= 2.4 layout
    from torch.amp import custom_bwd, custom_fwd

    def _fwd_dec(f):
        return custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)(f)

    def _bwd_dec(f):
        return custom_bwd(device_type="cuda")(f)
except ImportError:  # pragma: no cover
    from torch.cuda.amp import custom_bwd, custom_fwd

    def _fwd_dec(f):
        return custom_fwd(cast_inputs=torch.bfloat16)(f)

    def _bwd_dec(f):
        return custom_bwd(f)


class _LinearReLU(torch.autograd.Function):
    @staticmethod
    @_fwd_dec
    def forward(ctx, x, weight, bias):
        out = torch._addmm_activation(bias, x, weight.t(), use_gelu=False)
        ctx.save_for_backward(x, weight, out)
        return out

    @staticmethod
    @_bwd_dec
    def backward(ctx, gout):
        x, weight, out = ctx.saved_tensors
        g = gout * (out > 0)
        grad_x = g.mm(weight)
        grad_w = g.t().mm(x)
        grad_b = g.sum(0)
        return grad_x, grad_w, grad_b


class FusedLinearReLU(nn.Module):
    """Drop-in for ``nn.Linear(in_f, out_f)`` followed by ``nn.ReLU``.

    Parameter names match ``nn.Linear`` (``weight``/``bias``) so state dicts
    keyed by the position in an ``nn.Sequential`` stay compatible with a
    Linear+ReLU pair occupying one slot.
    """

    def __init__(self, in_features: int, out_features: int):
        super().__init__()
        self.in_features = int(in_features)
        self.out_features = int(out_features)
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features))
        nn.init.xavier_normal_(self.weight)
        nn.init.normal_(self.bias, std=math.sqrt(1.0 / out_features))

    def extra_repr(self) -> str:
        return f"in_features={self.in_features}, out_features={self.out_features}"

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            return _LinearReLU.apply(x, self.weight, self.bias)
        return torch.relu(torch.nn.functional.linear(x, self.weight, self.bias))
