"""Checkpoint contract + CPU offload + all-modes-simultaneously tests.

Parity targets: reference get/set_weights round-trip (dist_model_parallel.py
:971-1162), np.load(mmap) path (:911,919,950), offload tests
(dist_model_parallel_test.py:533-556), the 41-table all-modes test (:513-531).
"""

import numpy as np
import pytest
import torch

from conftest import run_distributed


def _round_trip_worker(rank, world, use_paths, tmpdir):
    import distributed_embeddings_amd as de
    sizes = [37, 110, 259, 16]
    tables = [de.TableConfig(s, 8) for s in sizes]
    model = de.DistributedEmbedding(tables, strategy="memory_balanced",
                                    column_slice_threshold=110 * 8 - 1)
    g = torch.Generator().manual_seed(5)
    weights = [torch.randn(s, 8, generator=g).numpy() for s in sizes]
    if use_paths:
        paths = []
        for t, w in enumerate(weights):
            p = f"{tmpdir}/w{t}.npy"
            if rank == 0:
                np.save(p, w)
            paths.append(p)
        import torch.distributed as dist
        dist.barrier()
        model.set_weights(paths)
    else:
        model.set_weights(weights)
    got = model.get_weights(all_ranks=True)
    errs = [float(np.abs(got[t] - weights[t]).max()) for t in range(len(sizes))]
    return errs


@pytest.mark.parametrize("use_paths", [False, True])
def test_get_set_weights_round_trip_world2(tmp_path, use_paths):
    results = run_distributed(_round_trip_worker, world=2,
                              args=(use_paths, str(tmp_path)))
    for rank_errs in results:
        assert max(rank_errs) < 1e-6


def test_get_set_weights_world1():
    import distributed_embeddings_amd as de
    sizes = [37, 110]
    model = de.DistributedEmbedding([de.TableConfig(s, 8) for s in sizes])
    weights = [np.random.RandomState(1).randn(s, 8).astype(np.float32) for s in sizes]
    model.set_weights(weights)
    got = model.get_weights()
    for w, g in zip(weights, got):
        assert np.allclose(w, g)


def test_cpu_offload_world1():
    import distributed_embeddings_amd as de
    sizes = [100, 5000, 50]
    model = de.DistributedEmbedding(
        [de.TableConfig(s, 8) for s in sizes],
        gpu_embedding_size=(100 + 50) * 8)  # largest table offloads
    offloaded = [getattr(l, "_cpu_offload", False) for l in model.col_layers]
    assert any(offloaded)
    ids = [torch.randint(0, s, (4,)) for s in sizes]
    outs = model(ids)
    assert len(outs) == 3
    weights = model.get_weights()
    for t, s in enumerate(sizes):
        assert weights[t].shape == (s, 8)


def _all_modes_worker(rank, world):
    import distributed_embeddings_amd as de
    # dp (tiny) + col (medium, width-sliced) + row (huge) simultaneously
    sizes = [8, 30, 400, 2000, 12, 900]
    tables = [de.TableConfig(s, 16) for s in sizes]
    model = de.DistributedEmbedding(
        tables, strategy="memory_balanced",
        data_parallel_threshold=30 * 16,
        row_slice_threshold=900 * 16,
        column_slice_threshold=400 * 16 - 1)
    g = torch.Generator().manual_seed(7)
    weights = [torch.randn(s, 16, generator=g).numpy() for s in sizes]
    model.set_weights(weights)
    ginp = torch.Generator().manual_seed(100)
    inputs = [torch.randint(0, s, (world * 4,), generator=ginp) for s in sizes]
    local = [x[rank * 4:(rank + 1) * 4] for x in inputs]
    outs = model(local)
    errs = []
    for t in range(len(sizes)):
        ref = torch.from_numpy(weights[t])[inputs[t][rank * 4:(rank + 1) * 4]]
        errs.append(float((outs[t] - ref).abs().max()))
    # groups actually exercised all three modes
    plan = model.strategy
    assert plan.dp_table_ids and plan.col_table_ids and plan.row_table_ids
    got = model.get_weights(all_ranks=True)
    werrs = [float(np.abs(got[t] - weights[t]).max()) for t in range(len(sizes))]
    return max(errs), max(werrs)


def test_all_modes_simultaneously_world2():
    results = run_distributed(_all_modes_worker, world=2)
    for fwd_err, w_err in results:
        assert fwd_err < 1e-5
        assert w_err < 1e-6


def _shared_multihot_worker(rank, world):
    import distributed_embeddings_amd as de
    # shared table: inputs 0 (hot1) and 1 (hot3) both map to table 0
    tables = [de.TableConfig(50, 8, "sum"), de.TableConfig(60, 8, "sum")]
    model = de.DistributedEmbedding(tables, input_table_map=[0, 0, 1])
    g = torch.Generator().manual_seed(3)
    weights = [torch.randn(50, 8, generator=g).numpy(),
               torch.randn(60, 8, generator=g).numpy()]
    model.set_weights(weights)
    gi = torch.Generator().manual_seed(11)
    inp0 = torch.randint(0, 50, (world * 4, 1), generator=gi)
    inp1 = torch.randint(0, 50, (world * 4, 3), generator=gi)
    inp2 = torch.randint(0, 60, (world * 4, 2), generator=gi)
    sl = slice(rank * 4, (rank + 1) * 4)
    outs = model([inp0[sl], inp1[sl], inp2[sl]])
    w0 = torch.from_numpy(weights[0])
    w1 = torch.from_numpy(weights[1])
    refs = [w0[inp0[sl]].sum(1), w0[inp1[sl]].sum(1), w1[inp2[sl]].sum(1)]
    return [float((o - r).abs().max()) for o, r in zip(outs, refs)]


def test_shared_tables_multihot_world2():
    results = run_distributed(_shared_multihot_worker, world=2)
    for errs in results:
        assert max(errs) < 1e-5


def test_bf16_table_dtype_module():
    import distributed_embeddings_amd as de
    model = de.DistributedEmbedding(
        [de.TableConfig(40, 8), de.TableConfig(60, 8, "sum")],
        table_dtype=torch.bfloat16)
    assert model.col_layers[0].weight.dtype == torch.bfloat16
    weights = [np.random.RandomState(0).randn(40, 8).astype(np.float32),
               np.random.RandomState(1).randn(60, 8).astype(np.float32)]
    model.set_weights(weights)
    outs = model([torch.randint(0, 40, (4,)),
                  torch.randint(0, 60, (4, 3))])
    assert all(torch.isfinite(o.float()).all() for o in outs)
    got = model.get_weights()
    assert np.allclose(got[0], weights[0], atol=0.01)  # bf16 rounding


def test_checkpoint_roundtrip_property():
    """Property: set_weights(get_weights()) is identity for random plans."""
    from hypothesis import given, settings, strategies as st
    import distributed_embeddings_amd as de

    @settings(max_examples=25, deadline=None)
    @given(st.lists(st.integers(2, 300), min_size=1, max_size=6),
           st.sampled_from([8, 16, 32]),
           st.sampled_from(["basic", "memory_balanced", "memory_optimized"]))
    def check(sizes, width, strategy):
        model = de.DistributedEmbedding(
            [de.TableConfig(s, width) for s in sizes], strategy=strategy)
        rng = np.random.RandomState(0)
        weights = [rng.randn(s, width).astype(np.float32) for s in sizes]
        model.set_weights(weights)
        got = model.get_weights()
        for w, g in zip(weights, got):
            assert np.array_equal(w, g)

    check()


def test_bf16_get_weights_all_modes():
    """get_weights must work for bf16 tables in every placement mode
    (bf16 tensors have no numpy dtype — conversion must go through fp32)."""
    import distributed_embeddings_amd as de
    sizes = [8, 400, 900]
    model = de.DistributedEmbedding(
        [de.TableConfig(s, 16) for s in sizes],
        data_parallel_threshold=8 * 16, row_slice_threshold=900 * 16,
        table_dtype=torch.bfloat16)
    plan = model.strategy
    assert plan.dp_table_ids and plan.col_table_ids and plan.row_table_ids
    weights = [np.random.RandomState(t).randn(s, 16).astype(np.float32)
               for t, s in enumerate(sizes)]
    model.set_weights(weights)
    got = model.get_weights()
    for w, g in zip(weights, got):
        assert g.dtype == np.float32
        assert np.allclose(w, g, rtol=0.01, atol=0.01)  # bf16 rounding


def _get_weights_rank0_worker(rank, world):
    import distributed_embeddings_amd as de
    sizes = [30, 200]
    model = de.DistributedEmbedding([de.TableConfig(s, 8) for s in sizes])
    g = torch.Generator().manual_seed(2)
    weights = [torch.randn(s, 8, generator=g).numpy() for s in sizes]
    model.set_weights(weights)
    got = model.get_weights(all_ranks=False)  # collective: all ranks call
    if rank != 0:
        return 0.0
    return max(float(np.abs(got[t] - weights[t]).max()) for t in range(2))


def test_get_weights_rank0_only_world2():
    """all_ranks=False is still a collective — every rank participates, no
    deadlock, rank 0 gets the full tables."""
    results = run_distributed(_get_weights_rank0_worker, world=2)
    assert results[0] < 1e-6


def _ckpt_dir_worker(rank, world, tmpdir):
    """save_embedding_checkpoint -> load into a DIFFERENTLY-sharded model
    (world stays, strategy changes) == original tables."""
    import distributed_embeddings_amd as de
    sizes = [50, 700, 33]
    tables = [de.TableConfig(s, 8, "sum") for s in sizes]
    m1 = de.DistributedEmbedding(tables, strategy="basic",
                                 row_slice_threshold=5000,
                                 data_parallel_threshold=300)
    g = torch.Generator().manual_seed(23)
    weights = [torch.randn(s, 8, generator=g) for s in sizes]
    m1.set_weights([w.numpy() for w in weights])
    de.save_embedding_checkpoint(m1, tmpdir, chunk_elements=64)
    m2 = de.DistributedEmbedding(tables, strategy="memory_balanced")
    de.load_embedding_checkpoint(m2, tmpdir, chunk_elements=64)
    out = m2.get_weights(all_ranks=True)
    return [torch.as_tensor(w) for w in out]


def test_checkpoint_dir_roundtrip_world2(tmp_path):
    outs = run_distributed(_ckpt_dir_worker, world=2, args=(str(tmp_path),))
    g = torch.Generator().manual_seed(23)
    expect = [torch.randn(s, 8, generator=g) for s in [50, 700, 33]]
    for o in outs:
        for got, want in zip(o, expect):
            assert torch.equal(got, want)


def test_checkpoint_dir_roundtrip_world1(tmp_path):
    out = _ckpt_dir_worker(0, 1, str(tmp_path))
    g = torch.Generator().manual_seed(23)
    for got, want in zip(out, [torch.randn(s, 8, generator=g)
                               for s in [50, 700, 33]]):
        assert torch.equal(got, want)
