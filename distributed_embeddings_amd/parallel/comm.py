"""Autograd-aware collectives over ``torch.distributed`` (RCCL / gloo).

MI355X-native replacement for the reference's Horovod backend
(``/root/reference/distributed_embeddings/python/layers/dist_model_parallel.py:22-24``
and call sites C1-C10 in SURVEY.md §2.2).  On ROCm the ``"nccl"`` backend IS
RCCL over xGMI: the 8-GPU node is a full point-to-point mesh (7 links/GPU), so
single fused all-to-alls with per-peer splits are the natural primitive — every
peer pair has a dedicated direct link.

Gradient-defined collectives:

* :func:`all_to_all_single` — grad is the reverse all-to-all with swapped
  splits (parity: Horovod alltoall autodiff used at C1-C3).
* :func:`all_gather`        — grad is reduce-scatter.
* :func:`reduce_scatter`    — grad is all-gather, deliberately **unscaled**
  (parity: ``grouped_reducescatter_unscaled``, reference ``:291-298``).

All functions are world_size==1 passthrough-safe and work on both the RCCL
("nccl") and gloo backends.
"""

import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist


def _strict_check(t: torch.Tensor):
    """Audit mode (DE_COMM_DEVICE=cuda): fail loudly when a CPU tensor would
    enter a collective.  On an 8-GPU RCCL run such a tensor aborts the job
    deep inside NCCL; this surfaces it in the world-2 gloo+CUDA test lane
    (RCCL refuses 2 ranks on one device — see tools/probe_rccl2.py — so this
    audit is how single-GPU leases validate device placement)."""
    if os.environ.get("DE_COMM_DEVICE") == "cuda" and not t.is_cuda:
        raise RuntimeError(
            f"CPU tensor (shape {tuple(t.shape)}, dtype {t.dtype}) entering a "
            "collective under DE_COMM_DEVICE=cuda — this would fail on RCCL")


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def world_size(group=None) -> int:
    if not is_initialized():
        return 1
    return dist.get_world_size(group)


def rank(group=None) -> int:
    if not is_initialized():
        return 0
    return dist.get_rank(group)


def barrier(group=None):
    if is_initialized():
        dist.barrier(group)


def _as_list(x):
    return None if x is None else [int(v) for v in x]


def backend_device(group=None) -> torch.device:
    """Device collectives must run on: CUDA for the nccl(RCCL) backend, CPU
    for gloo / uninitialized."""
    if not is_initialized():
        return torch.device("cpu")
    if dist.get_backend(group) == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def all_gather_ints(vals: Sequence[int], group=None) -> List[List[int]]:
    """All-gather one equal-length int vector per rank via tensor collectives.

    RCCL-safe replacement for ``all_gather_object`` (which pickles through
    CPU staging buffers and is neither graph-capturable nor portable across
    backends).  Every rank must pass the SAME number of ints.
    """
    w = world_size(group)
    if w == 1:
        return [[int(v) for v in vals]]
    dev = backend_device(group)
    t = torch.tensor([int(v) for v in vals], dtype=torch.long, device=dev)
    out = torch.empty(w * t.numel(), dtype=torch.long, device=dev)
    dist.all_gather_into_tensor(out, t, group=group)
    return out.view(w, -1).cpu().tolist()


def all_gather_int_vectors(vals: Sequence[int], group=None) -> List[List[int]]:
    """All-gather per-rank int vectors of DIFFERING lengths (two tensor
    collectives: lengths, then max-padded payload)."""
    w = world_size(group)
    if w == 1:
        return [[int(v) for v in vals]]
    lens = [row[0] for row in all_gather_ints([len(vals)], group)]
    mx = max(lens + [1])
    dev = backend_device(group)
    t = torch.zeros(mx, dtype=torch.long, device=dev)
    if len(vals):
        t[:len(vals)] = torch.tensor([int(v) for v in vals], dtype=torch.long)
    out = torch.empty(w * mx, dtype=torch.long, device=dev)
    dist.all_gather_into_tensor(out, t, group=group)
    rows = out.view(w, mx).cpu().tolist()
    return [row[:k] for row, k in zip(rows, lens)]


class _AllToAllSingle(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        _strict_check(inp)
        # execute on the backend's device (no-op copy on the matched path:
        # nccl+cuda or gloo+cpu); restore to the caller's device
        dev = backend_device(group)
        src = inp.contiguous().to(dev)
        n_out = sum(out_splits) if out_splits is not None else inp.shape[0]
        out = src.new_empty((n_out,) + tuple(inp.shape[1:]))
        dist.all_to_all_single(out, src,
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits, group=group)
        return out.to(inp.device)

    @staticmethod
    def backward(ctx, grad_out):
        _strict_check(grad_out)
        dev = backend_device(ctx.group)
        src = grad_out.contiguous().to(dev)
        n_in = sum(ctx.in_splits) if ctx.in_splits is not None else grad_out.shape[0]
        grad_in = src.new_empty((n_in,) + tuple(grad_out.shape[1:]))
        dist.all_to_all_single(grad_in, src,
                               output_split_sizes=ctx.in_splits,
                               input_split_sizes=ctx.out_splits, group=ctx.group)
        return grad_in.to(grad_out.device), None, None, None


def all_to_all_single(
    inp: torch.Tensor,
    output_split_sizes: Optional[Sequence[int]] = None,
    input_split_sizes: Optional[Sequence[int]] = None,
    group=None,
) -> torch.Tensor:
    """Fused all-to-all along dim 0 with optional uneven per-peer splits."""
    if world_size(group) == 1:
        return inp
    return _AllToAllSingle.apply(inp, _as_list(output_split_sizes),
                                 _as_list(input_split_sizes), group)


class _AllGather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, group):
        ctx.group = group
        ctx.in_rows = inp.shape[0]
        _strict_check(inp)
        dev = backend_device(group)
        src = inp.contiguous().to(dev)
        w = world_size(group)
        out = src.new_empty((inp.shape[0] * w,) + tuple(inp.shape[1:]))
        dist.all_gather_into_tensor(out, src, group=group)
        return out.to(inp.device)

    @staticmethod
    def backward(ctx, grad_out):
        _strict_check(grad_out)
        dev = backend_device(ctx.group)
        src = grad_out.contiguous().to(dev)
        grad_in = src.new_empty((ctx.in_rows,) + tuple(grad_out.shape[1:]))
        dist.reduce_scatter_tensor(grad_in, src, group=ctx.group)
        return grad_in.to(grad_out.device), None


def all_gather(inp: torch.Tensor, group=None) -> torch.Tensor:
    """Gathers equal-shaped tensors from all ranks along dim 0 (grad = reduce-scatter)."""
    if world_size(group) == 1:
        return inp
    return _AllGather.apply(inp, group)


class _ReduceScatter(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, group):
        ctx.group = group
        _strict_check(inp)
        dev = backend_device(group)
        src = inp.contiguous().to(dev)
        w = world_size(group)
        out = src.new_empty((inp.shape[0] // w,) + tuple(inp.shape[1:]))
        dist.reduce_scatter_tensor(out, src, group=group)
        return out.to(inp.device)

    @staticmethod
    def backward(ctx, grad_out):
        _strict_check(grad_out)
        dev = backend_device(ctx.group)
        src = grad_out.contiguous().to(dev)
        w = world_size(ctx.group)
        grad_in = src.new_empty((grad_out.shape[0] * w,) + tuple(grad_out.shape[1:]))
        dist.all_gather_into_tensor(grad_in, src, group=ctx.group)
        return grad_in.to(grad_out.device), None


def reduce_scatter(inp: torch.Tensor, group=None) -> torch.Tensor:
    """Sum-reduce-scatter along dim 0; grad is all-gather (unscaled).

    Parity: reference ``grouped_reducescatter_unscaled``
    (dist_model_parallel.py:291-298).
    """
    if world_size(group) == 1:
        return inp
    return _ReduceScatter.apply(inp, group)


def all_gather_uneven(inp: torch.Tensor, group=None) -> List[torch.Tensor]:
    """All-gather of per-rank tensors with differing dim-0 sizes.

    Used by the checkpoint (get_weights) path — parity with reference chunked
    ``hvd.allgather`` (dist_model_parallel.py:1084-1089).
    """
    w = world_size(group)
    if w == 1:
        return [inp]
    sizes = [row[0] for row in all_gather_ints([int(inp.shape[0])], group)]
    mx = max(sizes)
    # stage through the backend's device (nccl refuses CPU tensors), return
    # on the caller's device
    dev = backend_device(group)
    padded = inp.to(dev)
    if inp.shape[0] < mx:
        padded = torch.cat(
            [padded, padded.new_zeros((mx - inp.shape[0],) + tuple(inp.shape[1:]))])
    out = [padded.new_empty((mx,) + tuple(inp.shape[1:])) for _ in range(w)]
    dist.all_gather(out, padded.contiguous(), group=group)
    return [o[:s].to(inp.device) for o, s in zip(out, sizes)]


def broadcast(t: torch.Tensor, src: int = 0, group=None) -> torch.Tensor:
    if world_size(group) > 1:
        dev = backend_device(group)
        if t.device != dev:
            # cold-path utility: stage through the backend's device (e.g. a
            # CPU-offloaded table broadcast on an nccl job)
            staged = t.to(dev)
            dist.broadcast(staged, src=src, group=group)
            t.copy_(staged)
        else:
            dist.broadcast(t, src=src, group=group)
    return t


def allreduce_sum_(t: torch.Tensor, group=None) -> torch.Tensor:
    if world_size(group) > 1:
        dev = backend_device(group)
        if t.device != dev:
            staged = t.to(dev)
            dist.all_reduce(staged, op=dist.ReduceOp.SUM, group=group)
            t.copy_(staged)
        else:
            dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    return t
