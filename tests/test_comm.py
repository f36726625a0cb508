"""Direct gradient tests for the autograd-aware collectives (gloo world=2).

Parity targets: Horovod alltoall autodiff (C1-C3) and
``grouped_reducescatter_unscaled``'s allgather gradient (reference
dist_model_parallel.py:291-298).
"""

import torch

from conftest import run_distributed


def _a2a_grad_worker(rank, world):
    from distributed_embeddings_amd.parallel import comm
    # rank r sends [r*10+0 .. r*10+5]; splits 4/2 out, so grads must route back
    x = (torch.arange(6, dtype=torch.float32) + rank * 10).requires_grad_(True)
    in_splits = [4, 2]
    # out sizes mirror the peers' in-splits toward me
    out_splits = [4, 4] if rank == 0 else [2, 2]
    y = comm.all_to_all_single(x, out_splits, in_splits)
    w = torch.arange(y.numel(), dtype=torch.float32) + 1 + rank * 100
    (y * w).sum().backward()
    return {"y": y.detach(), "gx": x.grad}


def test_all_to_all_grad_world2():
    results = run_distributed(_a2a_grad_worker, world=2)
    # forward: rank0 receives [r0[0:4], r1[0:4]] -> [0,1,2,3,10,11,12,13]
    assert torch.equal(results[0]["y"], torch.tensor([0., 1, 2, 3, 10, 11, 12, 13]))
    assert torch.equal(results[1]["y"], torch.tensor([4., 5, 14, 15]))
    # grad of x on rank0: first 4 elems got rank0's w[0:4]=[1..4],
    # last 2 went to rank1 where they were positions 0:2 with w=[101,102]
    assert torch.equal(results[0]["gx"], torch.tensor([1., 2, 3, 4, 101, 102]))
    # rank1: first 4 elems -> rank0 positions 4:8 (w=[5..8]); last 2 stay
    # on rank1 positions 2:4 (w=[103,104])
    assert torch.equal(results[1]["gx"], torch.tensor([5., 6, 7, 8, 103, 104]))


def _allgather_grad_worker(rank, world):
    from distributed_embeddings_amd.parallel import comm
    x = torch.full((3, 2), float(rank + 1)).requires_grad_(True)
    y = comm.all_gather(x)          # [world*3, 2]
    w = torch.arange(world * 6, dtype=torch.float32).view(world * 3, 2)
    (y * w).sum().backward()
    return {"y": y.detach(), "gx": x.grad}


def test_all_gather_grad_world2():
    results = run_distributed(_allgather_grad_worker, world=2)
    w = torch.arange(12, dtype=torch.float32).view(6, 2)
    for rank in range(2):
        # forward gathers both ranks' blocks
        assert torch.equal(results[rank]["y"][:3], torch.full((3, 2), 1.0))
        assert torch.equal(results[rank]["y"][3:], torch.full((3, 2), 2.0))
        # grad = reduce-scatter(w) = my block of w summed over ranks (each
        # rank contributes identical w, so my slice times world)
        expect = w[rank * 3:(rank + 1) * 3] * 2
        assert torch.equal(results[rank]["gx"], expect)


def _rs_grad_worker(rank, world):
    from distributed_embeddings_amd.parallel import comm
    x = (torch.arange(4, dtype=torch.float32) + rank * 10).requires_grad_(True)
    y = comm.reduce_scatter(x)      # my half of the sum over ranks
    w = torch.tensor([3., 5]) + rank
    (y * w).sum().backward()
    return {"y": y.detach(), "gx": x.grad}


def test_reduce_scatter_unscaled_grad_world2():
    results = run_distributed(_rs_grad_worker, world=2)
    # forward: sum over ranks = [0+10, 1+11, 2+12, 3+13]; rank r gets half r
    assert torch.equal(results[0]["y"], torch.tensor([10., 12]))
    assert torch.equal(results[1]["y"], torch.tensor([14., 16]))
    # grad = UNSCALED all-gather of per-rank upstreams: [w0, w1] on every rank
    for rank in range(2):
        assert torch.equal(results[rank]["gx"], torch.tensor([3., 5, 4, 6]))
