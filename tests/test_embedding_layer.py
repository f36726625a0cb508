"""Layer-level tests for Embedding / ConcatOneHotEmbedding.

Mirrors reference ``tests/embedding_test.py``: 1D/2D/3D dense, sum/mean,
ragged, sparse, grads vs reference under Adagrad, int32 ids.
"""

import pytest
import torch

from distributed_embeddings_amd import ConcatOneHotEmbedding, Embedding, Ragged


def test_dense_1d(seed):
    e = Embedding(20, 4)
    ids = torch.randint(0, 20, (7,))
    assert e(ids).shape == (7, 4)
    assert torch.equal(e(ids), e.weight[ids])


def test_dense_2d_no_combiner(seed):
    e = Embedding(20, 4)
    ids = torch.randint(0, 20, (7, 3))
    assert e(ids).shape == (7, 3, 4)


def test_dense_3d_combiner(seed):
    e = Embedding(20, 4, combiner="mean")
    ids = torch.randint(0, 20, (2, 7, 3))
    out = e(ids)
    assert out.shape == (2, 7, 4)
    ref = e.weight[ids].mean(dim=2)
    assert torch.allclose(out, ref, atol=1e-6)


def test_int32_ids(seed):
    e = Embedding(20, 4, combiner="sum")
    ids = torch.randint(0, 20, (5, 2), dtype=torch.int32)
    assert e(ids).shape == (5, 4)


def test_ragged(seed):
    e = Embedding(20, 4, combiner="sum")
    r = Ragged.from_lists([[1, 2], [3], [4, 5, 6]])
    out = e(r)
    assert out.shape == (3, 4)
    assert torch.allclose(out[1], e.weight[3])


def test_sparse(seed):
    e = Embedding(20, 4, combiner="sum")
    indices = torch.tensor([[0, 0], [1, 0], [1, 1]]).t()
    sp = torch.sparse_coo_tensor(indices, torch.tensor([5, 6, 7]), (2, 2))
    out = e(sp)
    assert torch.allclose(out[1], e.weight[6] + e.weight[7], atol=1e-6)


def test_adagrad_step_matches_oracle(seed):
    """One Adagrad step on the custom path == on a plain-gather oracle
    (parity: reference embedding_test.py grads-under-Adagrad tests)."""
    ids = torch.randint(0, 30, (16, 4))
    w0 = torch.randn(30, 8)

    e = Embedding(30, 8, combiner="sum")
    with torch.no_grad():
        e.weight.copy_(w0)
    opt = torch.optim.Adagrad(e.parameters(), lr=0.1)
    out = e(ids)
    out.square().sum().backward()
    # sparse grads: Adagrad supports them natively
    opt.step()

    w_ref = w0.clone().requires_grad_(True)
    opt_ref = torch.optim.Adagrad([w_ref], lr=0.1)
    out_ref = w_ref[ids].sum(1)
    out_ref.square().sum().backward()
    opt_ref.step()

    assert torch.allclose(e.weight, w_ref, atol=1e-6)


def test_rejects_1d_with_combiner():
    e = Embedding(10, 2, combiner="sum")
    with pytest.raises(ValueError):
        e(torch.zeros(5, dtype=torch.long))


def test_oob_zero_mode(seed):
    e = Embedding(10, 4)
    e._oob_zero = True
    ids = torch.tensor([0, 11, -3, 9])
    out = e(ids)
    assert torch.equal(out[1], torch.zeros(4))
    assert torch.equal(out[2], torch.zeros(4))
    assert torch.equal(out[3], e.weight[9])


def test_concat_onehot(seed):
    layer = ConcatOneHotEmbedding([5, 7, 3], 4)
    ids = torch.tensor([[1, 2, 0], [4, 6, 2]])
    out = layer(ids)
    assert out.shape == (2, 3, 4)
    assert torch.equal(out[0, 1], layer.weight[5 + 2])
    assert torch.equal(out[1, 2], layer.weight[5 + 7 + 2])


def test_sparse_embedding_optimizer_cpu_matches_torch(seed):
    """SparseEmbeddingOptimizer (cpu path) == torch.optim on same grads."""
    from distributed_embeddings_amd import Embedding, SparseEmbeddingOptimizer
    for method, torch_opt in [("sgd", lambda p: torch.optim.SGD(p, lr=0.1)),
                              ("adagrad", lambda p: torch.optim.Adagrad(p, lr=0.1, eps=1e-10))]:
        ids = torch.randint(0, 50, (16, 4))
        w0 = torch.randn(50, 8)
        e1 = Embedding(50, 8, combiner="sum")
        e2 = Embedding(50, 8, combiner="sum")
        with torch.no_grad():
            e1.weight.copy_(w0)
            e2.weight.copy_(w0)
        o1 = SparseEmbeddingOptimizer(e1.parameters(), lr=0.1, method=method)
        o2 = torch_opt(list(e2.parameters()))
        for _ in range(3):
            o1.zero_grad(); o2.zero_grad()
            e1(ids).square().sum().backward()
            e2(ids).square().sum().backward()
            o1.step(); o2.step()
        assert torch.allclose(e1.weight, e2.weight, atol=1e-5), method


def test_fused_sgd_cpu_matches_explicit(seed):
    """In-backward fused SGD == explicit sparse-grad SGD step (CPU path)."""
    from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer
    w0 = torch.randn(40, 8)
    lists = [[1, 2, 3], [2], [5, 5, 7], [0]]
    up = torch.randn(4, 8)
    for combiner in ("sum", "mean"):
        e1 = Embedding(40, 8, combiner=combiner)
        e2 = Embedding(40, 8, combiner=combiner)
        with torch.no_grad():
            e1.weight.copy_(w0); e2.weight.copy_(w0)
        e1.enable_fused_sgd(0.1)
        out1 = e1(Ragged.from_lists(lists))
        out1.backward(up)
        assert e1.weight.grad is None
        o2 = SparseEmbeddingOptimizer(e2.parameters(), lr=0.1)
        out2 = e2(Ragged.from_lists(lists))
        out2.backward(up)
        o2.step()
        assert torch.allclose(out1, out2, atol=1e-6)
        assert torch.allclose(e1.weight, e2.weight, atol=1e-6), combiner


def test_fused_adagrad_cpu_matches_explicit(seed):
    from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer
    w0 = torch.randn(40, 8)
    lists = [[1, 2, 3], [2], [5, 5, 7], [0]]
    up = torch.randn(4, 8)
    e1 = Embedding(40, 8, combiner="sum")
    e2 = Embedding(40, 8, combiner="sum")
    with torch.no_grad():
        e1.weight.copy_(w0); e2.weight.copy_(w0)
    e1.enable_fused_optimizer("adagrad", 0.1)
    o2 = SparseEmbeddingOptimizer(e2.parameters(), lr=0.1, method="adagrad")
    for _ in range(3):
        o2.zero_grad()
        out1 = e1(Ragged.from_lists(lists))
        out1.backward(up)
        out2 = e2(Ragged.from_lists(lists))
        out2.backward(up)
        o2.step()
    assert e1.weight.grad is None
    assert torch.allclose(e1.weight, e2.weight, atol=1e-5), \
        float((e1.weight - e2.weight).abs().max())


def test_bf16_table_forward_matches_fp32(seed):
    """bf16 table storage: forward accumulates fp32 and matches the fp32
    lookup of the (bf16-rounded) values exactly."""
    from distributed_embeddings_amd import Embedding, Ragged, embedding_lookup
    w32 = torch.randn(50, 16)
    wbf = w32.bfloat16()
    r = Ragged.from_lists([[1, 2, 3], [7], [4, 4, 9]])
    out_bf = embedding_lookup(wbf, r, "sum")
    out_ref = embedding_lookup(wbf.float(), r, "sum")
    assert out_bf.dtype == torch.float32
    assert torch.equal(out_bf, out_ref)


def test_bf16_table_training_cpu(seed):
    from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer
    e = Embedding(40, 8, combiner="sum", dtype=torch.bfloat16)
    opt = SparseEmbeddingOptimizer(e.parameters(), lr=0.1, method="adagrad")
    r = Ragged.from_lists([[1, 2], [3, 3], [5]])
    for _ in range(2):
        opt.zero_grad()
        out = e(r)
        assert out.dtype == torch.float32
        out.square().sum().backward()
        assert e.weight.grad.dtype == torch.bfloat16
        opt.step()
    assert e.weight.dtype == torch.bfloat16
    assert torch.isfinite(e.weight.float()).all()


def test_bf16_fused_optimizer_cpu(seed):
    from distributed_embeddings_amd import Embedding, Ragged
    for method in ("sgd", "adagrad"):
        e = Embedding(40, 8, combiner="sum", dtype=torch.bfloat16)
        e.enable_fused_optimizer(method, 0.1)
        assert e._fused_state.dtype == torch.float32
        w0 = e.weight.detach().clone()
        out = e(Ragged.from_lists([[1, 2], [3]]))
        out.sum().backward()
        assert e.weight.grad is None
        assert not torch.equal(e.weight, w0), method


def test_embedding_forward_property(seed):
    """Property test: random shapes/dtypes/combiners vs a plain oracle."""
    from hypothesis import given, settings, strategies as st
    from distributed_embeddings_amd import Embedding

    @settings(max_examples=60, deadline=None)
    @given(st.integers(1, 200), st.sampled_from([1, 3, 8, 17, 64, 96, 130]),
           st.sampled_from([None, "sum", "mean"]),
           st.integers(1, 12), st.integers(1, 6), st.booleans())
    def check(vocab, width, combiner, batch, hot, use_bf16):
        dtype = torch.bfloat16 if use_bf16 else torch.float32
        e = Embedding(vocab, width, combiner, dtype=dtype)
        if combiner is None:
            ids = torch.randint(0, vocab, (batch,))
            out = e(ids)
            ref = e.weight[ids]
            assert out.shape == (batch, width)
        else:
            ids = torch.randint(0, vocab, (batch, hot))
            out = e(ids)
            w = e.weight.float()
            ref = w[ids].sum(1) if combiner == "sum" else w[ids].mean(1)
            assert out.shape == (batch, width)
        assert torch.allclose(out.float(), ref.float(), atol=1e-4), \
            (vocab, width, combiner, batch, hot, use_bf16)

    check()


def test_sparse_optimizer_state_dict_roundtrip(seed):
    """Standard torch optimizer checkpointing: save/load state_dict mid-run
    must continue identically to an uninterrupted run (adagrad state)."""
    from distributed_embeddings_amd import Embedding
    from distributed_embeddings_amd.parallel.optim import SparseEmbeddingOptimizer

    def train(n_steps, reload_at=None):
        torch.manual_seed(seed)
        emb = Embedding(50, 8, combiner="sum")
        opt = SparseEmbeddingOptimizer(emb.parameters(), lr=0.1,
                                       method="adagrad")
        g = torch.Generator().manual_seed(3)
        for i in range(n_steps):
            if i == reload_at:
                sd_o, sd_m = opt.state_dict(), emb.state_dict()
                emb = Embedding(50, 8, combiner="sum")
                emb.load_state_dict(sd_m)
                opt = SparseEmbeddingOptimizer(emb.parameters(), lr=0.1,
                                               method="adagrad")
                opt.load_state_dict(sd_o)
            ids = torch.randint(0, 50, (16, 3), generator=g)
            opt.zero_grad()
            (emb(ids) ** 2).sum().backward()
            opt.step()
        return emb.weight.detach()

    w_plain = train(6)
    w_reload = train(6, reload_at=3)
    assert torch.allclose(w_plain, w_reload, atol=1e-6)


def test_fused_adagrad_state_in_state_dict(seed):
    """The fused optimizer's accumulator is a registered buffer, so module
    state_dict checkpointing resumes fused-Adagrad training exactly."""
    from distributed_embeddings_amd import Embedding

    def train(n_steps, reload_at=None):
        torch.manual_seed(seed)
        emb = Embedding(50, 8, combiner="sum")
        emb.enable_fused_optimizer("adagrad", lr=0.1)
        g = torch.Generator().manual_seed(5)
        for i in range(n_steps):
            if i == reload_at:
                sd = emb.state_dict()
                emb = Embedding(50, 8, combiner="sum")
                emb.enable_fused_optimizer("adagrad", lr=0.1)
                emb.load_state_dict(sd)
            ids = torch.randint(0, 50, (16, 3), generator=g)
            (emb(ids) ** 2).sum().backward()   # update applied in backward
        return emb.weight.detach()

    w_plain = train(6)
    w_reload = train(6, reload_at=3)
    assert torch.allclose(w_plain, w_reload, atol=1e-6)


def test_enable_fused_optimizer_reconfigure(seed):
    """enable_fused_optimizer may be called again (e.g. sgd -> adagrad)."""
    from distributed_embeddings_amd import Embedding
    emb = Embedding(20, 8, combiner="sum")
    emb.enable_fused_optimizer("sgd", 0.1)
    emb.enable_fused_optimizer("adagrad", 0.05)
    assert emb._fused_method == "adagrad"
    assert emb._fused_state.shape == emb.weight.shape
    ids = torch.randint(0, 20, (4, 2))
    (emb(ids) ** 2).sum().backward()
    assert (emb._fused_state > 0).any()


def test_fused_sgd_dense_input_matches_explicit(seed):
    """Dense [b, hotness] inputs must hit the fused update too (they are
    converted to CSR when a fused optimizer is enabled)."""
    from distributed_embeddings_amd import Embedding, SparseEmbeddingOptimizer
    w0 = torch.randn(40, 8)
    ids = torch.randint(0, 40, (6, 3), generator=torch.Generator().manual_seed(seed))
    up = torch.randn(6, 8)
    for combiner in ("sum", "mean"):
        e1 = Embedding(40, 8, combiner=combiner)
        e2 = Embedding(40, 8, combiner=combiner)
        with torch.no_grad():
            e1.weight.copy_(w0); e2.weight.copy_(w0)
        e1.enable_fused_sgd(0.1)
        out1 = e1(ids)
        out1.backward(up)
        assert e1.weight.grad is None
        o2 = SparseEmbeddingOptimizer(e2.parameters(), lr=0.1)
        out2 = e2(ids)
        out2.backward(up)
        o2.step()
        assert torch.allclose(out1, out2, atol=1e-6)
        assert torch.allclose(e1.weight, e2.weight, atol=1e-6), combiner


def test_fused_optimizer_property(seed):
    """Property: in-backward fused update == explicit sparse-optimizer step
    for random shapes, hotness, combiner and method (CPU oracle path)."""
    from hypothesis import given, settings, strategies as st
    from distributed_embeddings_amd import (Embedding, Ragged,
                                            SparseEmbeddingOptimizer)

    @settings(max_examples=30, deadline=None)
    @given(st.integers(0, 10_000), st.integers(2, 60), st.integers(1, 32),
           st.integers(1, 12), st.integers(1, 6),
           st.sampled_from(["sum", "mean"]), st.sampled_from(["sgd", "adagrad"]))
    def check(s, vocab, width, batch, hot, combiner, method):
        g = torch.Generator().manual_seed(s)
        w0 = torch.randn(vocab, width, generator=g)
        lengths = torch.randint(0, hot + 1, (batch,), generator=g)
        values = torch.randint(0, vocab, (int(lengths.sum()),), generator=g)
        ragged = Ragged.from_row_lengths(values, lengths)
        up = torch.randn(batch, width, generator=g)

        e1 = Embedding(vocab, width, combiner=combiner)
        e2 = Embedding(vocab, width, combiner=combiner)
        with torch.no_grad():
            e1.weight.copy_(w0); e2.weight.copy_(w0)
        e1.enable_fused_optimizer(method, 0.1)
        e1(ragged).backward(up)
        opt = SparseEmbeddingOptimizer(e2.parameters(), lr=0.1, method=method)
        e2(ragged).backward(up)
        opt.step()
        assert torch.allclose(e1.weight, e2.weight, atol=1e-5), \
            float((e1.weight - e2.weight).abs().max())

    check()
