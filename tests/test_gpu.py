"""GPU (MI355X) kernel numerics tests — every test compares the HIP kernels
against the plain-PyTorch fp32 reference of the same op.

Run via: gpurun -- python -m pytest tests -m gpu -x -q
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")


def _ragged_random(num_rows, vocab, max_hot, seed, device, empty_rows=True):
    from distributed_embeddings_amd import Ragged
    g = torch.Generator().manual_seed(seed)
    lengths = torch.randint(0 if empty_rows else 1, max_hot + 1, (num_rows,), generator=g)
    values = torch.randint(0, vocab, (int(lengths.sum()),), generator=g)
    return Ragged.from_row_lengths(values.to(device), lengths.to(device))


@requires_gpu
@pytest.mark.parametrize("width", [4, 8, 16, 32, 33, 48, 64, 96, 128, 200, 256, 512])
@pytest.mark.parametrize("combiner", ["sum", "mean"])
def test_csr_forward_vs_cpu(width, combiner):
    from distributed_embeddings_amd import embedding_lookup
    from distributed_embeddings_amd.ops.embedding_lookup import _csr_lookup_ref
    torch.manual_seed(width)
    vocab = 1000
    w = torch.randn(vocab, width)
    ragged = _ragged_random(257, vocab, 7, width, "cuda")
    out = embedding_lookup(w.cuda(), ragged, combiner)
    ref = _csr_lookup_ref(w, ragged.values.cpu(), ragged.row_splits.cpu(), combiner)
    assert torch.allclose(out.cpu(), ref, atol=1e-4), \
        f"max err {(out.cpu() - ref).abs().max()}"


@requires_gpu
def test_csr_forward_oob_and_empty():
    from distributed_embeddings_amd import Ragged, embedding_lookup
    w = torch.randn(10, 16).cuda()
    r = Ragged.from_lists([[1, 2], [], [-5, 3], [11, 4]], device="cuda")
    out = embedding_lookup(w, r, "sum")
    assert torch.equal(out[1].cpu(), torch.zeros(16))
    assert torch.allclose(out[2], w[3])
    assert torch.allclose(out[3], w[4])


@requires_gpu
@pytest.mark.parametrize("combiner", ["sum", "mean"])
@pytest.mark.parametrize("width", [16, 128])
def test_csr_backward_vs_cpu(width, combiner):
    from distributed_embeddings_amd import embedding_lookup
    torch.manual_seed(width + 1)
    vocab = 500
    w_cpu = torch.randn(vocab, width).requires_grad_(True)
    w_gpu = w_cpu.detach().clone().cuda().requires_grad_(True)
    ragged = _ragged_random(129, vocab, 9, width, "cuda")
    upstream = torch.randn(129, width)

    out_gpu = embedding_lookup(w_gpu, ragged, combiner)
    out_gpu.backward(upstream.cuda())
    out_cpu = embedding_lookup(w_cpu, ragged.to("cpu"), combiner)
    out_cpu.backward(upstream)

    g_gpu = w_gpu.grad.coalesce()
    g_cpu = w_cpu.grad.coalesce()
    assert torch.equal(g_gpu.indices().cpu(), g_cpu.indices())
    assert torch.allclose(g_gpu.values().cpu(), g_cpu.values(), atol=1e-4), \
        f"max err {(g_gpu.values().cpu() - g_cpu.values()).abs().max()}"


@requires_gpu
def test_csr_backward_power_law_skew():
    """Heavy dup ids (power-law) — stresses sort + segmented-sum path."""
    from distributed_embeddings_amd import Ragged, embedding_lookup
    from distributed_embeddings_amd.utils.input_gen import power_law_ids
    torch.manual_seed(3)
    vocab = 10000
    ids = power_law_ids(vocab, (20000,), alpha=1.2).cuda()
    splits = torch.arange(0, 20001, 4, device="cuda")  # 5000 rows x hotness 4
    w = torch.randn(vocab, 64).cuda().requires_grad_(True)
    out = embedding_lookup(w, Ragged(ids, splits), "mean")
    out.sum().backward()
    g = w.grad.coalesce()
    # CPU oracle
    w2 = w.detach().cpu().clone().requires_grad_(True)
    out2 = embedding_lookup(w2, Ragged(ids.cpu(), splits.cpu()), "mean")
    out2.sum().backward()
    g2 = w2.grad.coalesce()
    assert torch.equal(g.indices().cpu(), g2.indices())
    assert torch.allclose(g.values().cpu(), g2.values(), atol=1e-3)


@requires_gpu
def test_row_to_split_gpu():
    from distributed_embeddings_amd import row_to_split
    rows = torch.tensor([0, 0, 2, 2, 3], device="cuda")
    indices = torch.stack([rows, torch.zeros_like(rows)], dim=1)
    splits = row_to_split(indices, 4)
    assert splits.cpu().tolist() == [0, 2, 2, 4, 5]


@requires_gpu
def test_integer_lookup_gpu():
    from distributed_embeddings_amd import IntegerLookup
    lk = IntegerLookup(max_tokens=100).cuda()
    keys = torch.tensor([1000, 2000, 1000, 3000, 2000], device="cuda")
    out = lk(keys)
    o = out.cpu().tolist()
    # bijective assignment, dups consistent
    assert o[0] == o[2] and o[1] == o[4]
    assert len({o[0], o[1], o[3]}) == 3
    assert all(1 <= v <= 100 for v in (o[0], o[1], o[3]))
    # second batch: existing keys keep values
    out2 = lk(torch.tensor([3000, 1000], device="cuda"))
    assert out2.cpu().tolist() == [o[3], o[0]]
    # counts
    assert int(lk.counts[o[0]].cpu()) == 3


@requires_gpu
def test_integer_lookup_gpu_full_table_oov():
    from distributed_embeddings_amd import IntegerLookup
    lk = IntegerLookup(max_tokens=3).cuda()
    out = lk(torch.arange(100, 110, device="cuda"))
    vals = out.cpu().tolist()
    assert sorted(v for v in vals if v > 0) == [1, 2, 3]
    assert vals.count(0) == 7


@requires_gpu
def test_integer_lookup_gpu_large_batch_race():
    """Many duplicate new keys in one batch — insert race must stay consistent."""
    from distributed_embeddings_amd import IntegerLookup
    torch.manual_seed(0)
    lk = IntegerLookup(max_tokens=5000).cuda()
    keys = torch.randint(0, 3000, (100000,), device="cuda") * 7919
    out = lk(keys)
    # consistency: same key -> same value
    import collections
    m = {}
    for k, v in zip(keys.cpu().tolist(), out.cpu().tolist()):
        if k in m:
            assert m[k] == v, f"key {k} mapped to {m[k]} and {v}"
        m[k] = v
    assert len(set(m.values())) == len(m)


@requires_gpu
def test_embedding_layer_gpu_adagrad_step():
    from distributed_embeddings_amd import Embedding
    torch.manual_seed(5)
    ids = torch.randint(0, 300, (64, 4), device="cuda")
    w0 = torch.randn(300, 32)

    e = Embedding(300, 32, combiner="sum").cuda()
    with torch.no_grad():
        e.weight.copy_(w0)
    opt = torch.optim.Adagrad(e.parameters(), lr=0.1)
    e(ids).square().sum().backward()
    opt.step()

    w_ref = w0.clone().requires_grad_(True)
    opt_ref = torch.optim.Adagrad([w_ref], lr=0.1)
    w_ref[ids.cpu()].sum(1).square().sum().backward()
    opt_ref.step()
    assert torch.allclose(e.weight.cpu(), w_ref.detach(), atol=1e-4)


@requires_gpu
def test_dlrm_one_step():
    from distributed_embeddings_amd.models.dlrm import DLRM
    torch.manual_seed(6)
    sizes = [1000, 200, 5000, 33]
    model = DLRM(sizes, embedding_dim=64, num_numerical=13).cuda()
    b = 128
    num = torch.rand(b, 13, device="cuda")
    cats = [torch.randint(0, s, (b,), device="cuda") for s in sizes]
    labels = torch.randint(0, 2, (b, 1), device="cuda").float()
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            model(num, cats).float(), labels)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()


@requires_gpu
def test_native_extension_is_loaded():
    """The HIP extension must be the path that runs on GPU (no silent eager
    fallback)."""
    from distributed_embeddings_amd.ops import _backend
    assert _backend.available()
    ext = _backend.ops()
    assert "_hip_ops" in ext.__file__


@requires_gpu
@pytest.mark.parametrize("method", ["sgd", "adagrad"])
def test_sparse_optimizer_gpu_matches_cpu(method):
    from distributed_embeddings_amd import Embedding, SparseEmbeddingOptimizer
    torch.manual_seed(9)
    ids = torch.randint(0, 300, (64, 4))
    w0 = torch.randn(300, 128)
    e_gpu = Embedding(300, 128, combiner="sum").cuda()
    e_cpu = Embedding(300, 128, combiner="sum")
    with torch.no_grad():
        e_gpu.weight.copy_(w0)
        e_cpu.weight.copy_(w0)
    og = SparseEmbeddingOptimizer(e_gpu.parameters(), lr=0.1, method=method)
    oc = SparseEmbeddingOptimizer(e_cpu.parameters(), lr=0.1, method=method)
    for _ in range(3):
        og.zero_grad(); oc.zero_grad()
        e_gpu(ids.cuda()).square().sum().backward()
        e_cpu(ids).square().sum().backward()
        og.step(); oc.step()
    assert torch.allclose(e_gpu.weight.cpu(), e_cpu.weight, atol=1e-4), method


@requires_gpu
def test_long_segment_forward_vs_cpu():
    """Segments >> LONG_T exercise the two-kernel adaptive split path."""
    from distributed_embeddings_amd import Ragged, embedding_lookup
    from distributed_embeddings_amd.ops.embedding_lookup import _csr_lookup_ref
    torch.manual_seed(11)
    w = torch.randn(50, 128)
    # rows with lengths 1, 5000, 0, 300, 128, 129
    lens = [1, 5000, 0, 300, 128, 129]
    vals = torch.randint(0, 50, (sum(lens),))
    r = Ragged.from_row_lengths(vals.cuda(), torch.tensor(lens).cuda())
    for combiner in ("sum", "mean"):
        out = embedding_lookup(w.cuda(), r, combiner)
        ref = _csr_lookup_ref(w.double(), vals, Ragged.from_row_lengths(
            vals, torch.tensor(lens)).row_splits, combiner).float()
        # fp32 accumulation-order roundoff bound ~ len * eps * |max|
        assert torch.allclose(out.cpu(), ref, atol=5e-2), \
            f"{combiner}: {(out.cpu()-ref).abs().max()}"
    # narrow width too
    w16 = torch.randn(50, 16)
    out = embedding_lookup(w16.cuda(), r, "sum")
    ref = _csr_lookup_ref(w16, vals, Ragged.from_row_lengths(vals, torch.tensor(lens)).row_splits, "sum")
    assert torch.allclose(out.cpu(), ref, atol=1e-3)


@requires_gpu
def test_dot_interact_fused_vs_torch():
    """MFMA dot-interact vs the torch bmm oracle — asymmetric data so an
    operand/output transpose cannot pass (guide G9)."""
    from distributed_embeddings_amd.ops.dot_interact import (
        _DotInteract, _torch_dot_interact)
    torch.manual_seed(13)
    B, F, D = 512, 27, 128
    pad = 512
    feats = (torch.randn(B, F, D) * (1 + torch.arange(F).view(1, F, 1) * 0.1)
             ).bfloat16().cuda().requires_grad_(True)
    feats_ref = feats.detach().clone().requires_grad_(True)

    out = _DotInteract.apply(feats, pad)
    # fp32 oracle; outputs are bf16 so allow ~2 ulp relative (2^-7)
    ref = _torch_dot_interact(feats_ref.float(), pad)
    assert out.shape == ref.shape
    df = (out.float() - ref).abs()
    tol = ref.abs() * 2 ** -6 + 0.5
    assert bool((df <= tol).all()), \
        f"fwd err beyond bf16 ulp: max {(df - tol).max()}"

    gout = torch.randn_like(out)
    out.backward(gout)
    ref.backward(gout.float())
    gd = (feats.grad.float() - feats_ref.grad.float()).abs()
    gtol = feats_ref.grad.float().abs() * 2 ** -5 + 0.5
    assert bool((gd <= gtol).all()), \
        f"bwd err beyond bf16 ulp: max {(gd - gtol).max()}"


@requires_gpu
def test_dot_interact_small_shapes():
    from distributed_embeddings_amd.ops.dot_interact import (
        _DotInteract, _torch_dot_interact)
    for F, D in [(4, 32), (16, 64), (32, 128)]:
        feats = torch.randn(8, F, D).bfloat16().cuda()
        out = _DotInteract.apply(feats, F * (F - 1) // 2 + D)
        ref = _torch_dot_interact(feats.float(), out.shape[1])
        df = (out.float() - ref).abs()
        tol = ref.abs() * 2 ** -6 + 0.5
        assert bool((df <= tol).all()), f"F={F} D={D}"


@requires_gpu
def test_fused_sgd_gpu_matches_explicit():
    from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer
    torch.manual_seed(21)
    w0 = torch.randn(500, 128)
    ids = torch.randint(0, 500, (4096,), device="cuda")
    splits = torch.arange(0, 4097, 4, device="cuda")  # hotness 4
    up = torch.randn(1024, 128, device="cuda")
    e1 = Embedding(500, 128, combiner="sum").cuda()
    e2 = Embedding(500, 128, combiner="sum").cuda()
    with torch.no_grad():
        e1.weight.copy_(w0); e2.weight.copy_(w0)
    e1.enable_fused_sgd(0.1)
    out1 = e1(Ragged(ids, splits))
    out1.backward(up)
    assert e1.weight.grad is None
    o2 = SparseEmbeddingOptimizer(e2.parameters(), lr=0.1)
    out2 = e2(Ragged(ids, splits))
    out2.backward(up)
    o2.step()
    assert torch.allclose(out1, out2, atol=1e-5)
    # atomic scatter order differs from segmented sum: fp32 roundoff only
    assert torch.allclose(e1.weight, e2.weight, atol=1e-3), \
        float((e1.weight - e2.weight).abs().max())


@requires_gpu
def test_fused_adagrad_gpu_matches_explicit():
    from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer
    torch.manual_seed(31)
    w0 = torch.randn(500, 32)
    # heavy dups incl. tiny-vocab-style hot rows (long segments)
    ids = torch.cat([torch.randint(0, 500, (4000,)),
                     torch.zeros(2000, dtype=torch.long),
                     torch.full((2000,), 3)]).cuda()
    perm = torch.randperm(8000, device="cuda")
    ids = ids[perm]
    splits = torch.arange(0, 8001, 4, device="cuda")
    up = torch.randn(2000, 32, device="cuda")
    e1 = Embedding(500, 32, combiner="mean").cuda()
    e2 = Embedding(500, 32, combiner="mean").cuda()
    with torch.no_grad():
        e1.weight.copy_(w0); e2.weight.copy_(w0)
    e1.enable_fused_optimizer("adagrad", 0.1)
    o2 = SparseEmbeddingOptimizer(e2.parameters(), lr=0.1, method="adagrad")
    for _ in range(2):
        o2.zero_grad()
        e1(Ragged(ids, splits)).backward(up)
        e2(Ragged(ids, splits)).backward(up)
        o2.step()
    assert torch.allclose(e1.weight, e2.weight, atol=1e-3), \
        float((e1.weight - e2.weight).abs().max())


@requires_gpu
def test_distributed_embedding_module_gpu_world1():
    """Module-level GPU path: mixed hot1/multi-hot/ragged inputs through the
    fused groups, bf16 output_dtype, fused adagrad training step."""
    import distributed_embeddings_amd as de
    from distributed_embeddings_amd import Ragged
    torch.manual_seed(17)
    tables = [de.TableConfig(1000, 64, "sum"), de.TableConfig(50, 64, "sum"),
              de.TableConfig(300, 32, "mean"), de.TableConfig(77, 64, None)]
    with torch.device("cuda"):
        model = de.DistributedEmbedding(tables)
    g = torch.Generator().manual_seed(3)
    weights = [torch.randn(c.input_dim, c.output_dim, generator=g)
               for c in tables]
    model.set_weights([w.numpy() for w in weights])
    b = 64
    x0 = torch.randint(0, 1000, (b, 5), device="cuda")
    lens = torch.randint(0, 6, (b,), device="cuda")
    vals = torch.randint(0, 50, (int(lens.sum()),), device="cuda")
    x1 = Ragged.from_row_lengths(vals, lens)
    x2 = torch.randint(0, 300, (b, 3), device="cuda")
    x3 = torch.randint(0, 77, (b,), device="cuda")
    outs = model([x0, x1, x2, x3], output_dtype=torch.bfloat16)
    assert all(o.dtype == torch.bfloat16 for o in outs)
    refs = [weights[0][x0.cpu()].sum(1),
            None,
            weights[2][x2.cpu()].mean(1),
            weights[3][x3.cpu()]]
    for t in (0, 2, 3):
        err = (outs[t].float().cpu() - refs[t]).abs().max()
        assert float(err.detach()) < 0.5, f"table {t}: {err}"
    # ragged row check
    from distributed_embeddings_amd.ops.embedding_lookup import _csr_lookup_ref
    ref1 = _csr_lookup_ref(weights[1], vals.cpu(),
                           Ragged.from_row_lengths(vals.cpu(), lens.cpu()).row_splits,
                           "sum")
    assert float((outs[1].float().cpu() - ref1).abs().max()) < 0.5
    # fused adagrad training step runs end-to-end
    model.enable_fused_optimizer("adagrad", 0.05)
    outs = model([x0, x1, x2, x3])
    loss = sum(o.float().square().sum() for o in outs)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()


@requires_gpu
def test_sparse_coo_input_gpu():
    """torch sparse-COO id input path on GPU (row_to_split + CSR kernel)."""
    from distributed_embeddings_amd import embedding_lookup
    w = torch.randn(20, 16, device="cuda")
    indices = torch.tensor([[0, 0], [0, 1], [2, 0]], device="cuda").t()
    vals = torch.tensor([4, 7, 9], device="cuda")
    sp = torch.sparse_coo_tensor(indices, vals, (3, 2))
    out = embedding_lookup(w, sp, "sum")
    assert torch.allclose(out[0], w[4] + w[7], atol=1e-5)
    assert torch.equal(out[1].cpu(), torch.zeros(16))
    assert torch.allclose(out[2], w[9], atol=1e-5)


@requires_gpu
def test_hipgraph_capture_fused_train_step():
    """The fused-optimizer train step must be hipGraph-capturable (no host
    syncs anywhere in fwd+bwd)."""
    import distributed_embeddings_amd as de
    torch.manual_seed(23)
    with torch.device("cuda"):
        model = de.DistributedEmbedding(
            [de.TableConfig(5000, 64), de.TableConfig(300, 64)])
    model.enable_fused_optimizer("sgd", 0.01)
    ids0 = torch.randint(0, 5000, (256,), device="cuda")
    ids1 = torch.randint(0, 300, (256,), device="cuda")
    def step():
        outs = model([ids0, ids1], output_dtype=torch.bfloat16)
        loss = sum(o.float().square().sum() for o in outs)
        loss.backward()
    for _ in range(3):
        step()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step()
    w_before = model.col_layers[0].weight.detach().clone()
    g.replay()
    torch.cuda.synchronize()
    assert not torch.equal(w_before, model.col_layers[0].weight), \
        "replay did not update weights"


@requires_gpu
@pytest.mark.parametrize("width", [16, 32, 64, 128, 256])
def test_bf16_table_forward_gpu(width):
    """bf16 table storage on GPU == fp32 lookup of the bf16-rounded values."""
    from distributed_embeddings_amd import Ragged, embedding_lookup
    torch.manual_seed(width)
    wbf = torch.randn(500, width).bfloat16().cuda()
    r = _ragged_random(200, 500, 6, width, "cuda")
    out = embedding_lookup(wbf, r, "sum")
    ref = embedding_lookup(wbf.float(), r, "sum")
    assert out.dtype == torch.float32
    assert torch.allclose(out, ref, atol=1e-5), \
        f"w{width}: {(out - ref).abs().max()}"


@requires_gpu
def test_bf16_table_long_segments_gpu():
    from distributed_embeddings_amd import Ragged, embedding_lookup
    torch.manual_seed(2)
    wbf = torch.randn(20, 128).bfloat16().cuda()
    lens = torch.tensor([1, 3000, 0, 200])
    vals = torch.randint(0, 20, (int(lens.sum()),))
    r = Ragged.from_row_lengths(vals.cuda(), lens.cuda())
    out = embedding_lookup(wbf, r, "mean")
    ref = embedding_lookup(wbf.float(), r, "mean")
    assert torch.allclose(out, ref, atol=1e-2), (out - ref).abs().max()


@requires_gpu
@pytest.mark.parametrize("method", ["sgd", "adagrad"])
def test_bf16_fused_optimizer_gpu_matches_cpu(method):
    from distributed_embeddings_amd import Embedding, Ragged
    torch.manual_seed(5)
    w0 = torch.randn(300, 64).bfloat16()
    ids = torch.cat([torch.randint(0, 300, (2000,)),
                     torch.full((1500,), 7)])[torch.randperm(3500)]
    splits = torch.arange(0, 3501, 5)
    up = torch.randn(700, 64)
    e_g = Embedding(300, 64, combiner="sum", dtype=torch.bfloat16).cuda()
    e_c = Embedding(300, 64, combiner="sum", dtype=torch.bfloat16)
    with torch.no_grad():
        e_g.weight.copy_(w0)
        e_c.weight.copy_(w0)
    e_g.enable_fused_optimizer(method, 0.05)
    e_c.enable_fused_optimizer(method, 0.05)
    e_g(Ragged(ids.cuda(), splits.cuda())).backward(up.cuda())
    e_c(Ragged(ids, splits)).backward(up)
    d = (e_g.weight.detach().cpu().float() - e_c.weight.detach().float()).abs()
    # bf16 storage rounding: one step within a few ulp of the CPU oracle
    assert float(d.max()) < 0.05, float(d.max())


@requires_gpu
@pytest.mark.parametrize("method", ["sgd", "adagrad"])
def test_bf16_sparse_optimizer_gpu(method):
    from distributed_embeddings_amd import Embedding, Ragged, SparseEmbeddingOptimizer
    torch.manual_seed(6)
    e_g = Embedding(200, 32, combiner="sum", dtype=torch.bfloat16).cuda()
    e_c = Embedding(200, 32, combiner="sum", dtype=torch.bfloat16)
    with torch.no_grad():
        e_c.weight.copy_(e_g.weight.cpu())
    og = SparseEmbeddingOptimizer(e_g.parameters(), lr=0.05, method=method)
    oc = SparseEmbeddingOptimizer(e_c.parameters(), lr=0.05, method=method)
    r = Ragged.from_lists([[1, 2, 3], [5], [2, 9]])
    for _ in range(2):
        og.zero_grad(); oc.zero_grad()
        e_g(r.to("cuda")).square().sum().backward()
        e_c(r).square().sum().backward()
        og.step(); oc.step()
    d = (e_g.weight.detach().cpu().float() - e_c.weight.detach().float()).abs()
    assert float(d.max()) < 0.05, float(d.max())


@requires_gpu
def test_fused_sgd_dense_input_gpu_matches_cpu():
    """Dense [b, hotness] inputs route through the fused update on GPU too
    (converted to CSR in Embedding.forward)."""
    from distributed_embeddings_amd import Embedding
    torch.manual_seed(31)
    w0 = torch.randn(300, 64)
    ids = torch.randint(0, 300, (512, 4))
    up = torch.randn(512, 64)
    e_g = Embedding(300, 64, combiner="sum").cuda()
    e_c = Embedding(300, 64, combiner="sum")
    with torch.no_grad():
        e_g.weight.copy_(w0)
        e_c.weight.copy_(w0)
    e_g.enable_fused_sgd(0.1)
    e_c.enable_fused_sgd(0.1)
    out_g = e_g(ids.cuda())
    out_g.backward(up.cuda())
    out_c = e_c(ids)
    out_c.backward(up)
    assert e_g.weight.grad is None
    assert torch.allclose(out_g.cpu(), out_c, atol=1e-4)
    d = (e_g.weight.detach().cpu() - e_c.weight.detach()).abs()
    assert float(d.max()) < 1e-3, float(d.max())


@requires_gpu
def test_integer_lookup_gpu_auto_grow():
    """auto_grow=True: the hash_rehash kernel doubles capacity past the
    max_tokens guess; existing assignments and counts survive."""
    from distributed_embeddings_amd import IntegerLookup
    lk = IntegerLookup(max_tokens=4, auto_grow=True).cuda()
    a = lk(torch.tensor([5, 6, 7], device="cuda"))
    out = lk(torch.arange(1000, 1030, device="cuda"))
    assert lk.max_tokens >= 30
    assert (out.cpu() > 0).all()
    b = lk(torch.tensor([5, 6, 7], device="cuda"))
    assert torch.equal(a.cpu(), b.cpu())
    out2 = lk(torch.arange(1000, 1030, device="cuda"))
    assert torch.equal(out.cpu(), out2.cpu())
    # CPU/GPU interop after growth: the grown table probes identically on CPU
    c = lk.cpu()(torch.tensor([5, 6, 7]))
    assert torch.equal(a.cpu(), c)


@requires_gpu
def test_dot_interact_packed_vs_fallback():
    """Packed MFMA interaction kernel == torch oracle (fwd + both grads),
    including a non-trivial feature permutation."""
    from distributed_embeddings_amd.ops.dot_interact import (
        _torch_dot_interact, dot_interact_packed)
    torch.manual_seed(3)
    B, P, D = 512, 26, 128
    pad_to = 512
    perm_list = torch.randperm(P).tolist()
    perm = torch.tensor(perm_list, dtype=torch.int32, device="cuda")
    packed = torch.randn(P, B, D, device="cuda").bfloat16().requires_grad_(True)
    bottom = torch.randn(B, D, device="cuda").bfloat16().requires_grad_(True)

    out = dot_interact_packed(packed, bottom, perm, pad_to=pad_to)

    feats = torch.cat([bottom.detach().unsqueeze(0),
                       packed.detach()[perm.long()]]).transpose(0, 1).float()
    ref = _torch_dot_interact(feats.contiguous(), pad_to)
    assert out.shape == ref.shape
    err = (out.float() - ref).abs().max() / ref.abs().max()
    assert float(err) < 0.02, float(err)

    gout = torch.randn_like(out)
    out.backward(gout)
    f2 = torch.cat([bottom.detach().unsqueeze(0),
                    packed.detach()[perm.long()]]).transpose(0, 1)
    f2 = f2.contiguous().requires_grad_(True)
    ref2 = _torch_dot_interact(f2, pad_to)
    ref2.backward(gout)
    gb_ref = f2.grad[:, 0, :]
    errb = (bottom.grad.float() - gb_ref.float()).abs().max()
    assert float(errb) < 0.5, float(errb)
    # gpacked rows: feature f grad lands at packed row perm[f-1]
    gp_ref = f2.grad[:, 1:, :].transpose(0, 1)  # [P(feature order), B, D]
    inv = torch.empty(P, dtype=torch.long)
    inv[perm.long().cpu()] = torch.arange(P)
    gp_ref_packed = gp_ref[inv.cuda()]
    errp = (packed.grad.float() - gp_ref_packed.float()).abs().max()
    assert float(errp) < 0.5, float(errp)


@requires_gpu
def test_dlrm_packed_gpu_matches_legacy():
    """Whole-model check on GPU: packed interaction path == legacy path."""
    import os
    import distributed_embeddings_amd as de
    from distributed_embeddings_amd.models.dlrm import DLRM
    torch.manual_seed(0)
    sizes = [500, 600, 700, 400]
    with torch.device("cuda"):
        m = DLRM(sizes, embedding_dim=128, bottom_mlp_dims=(64, 128),
                 top_mlp_dims=(64, 1), num_numerical=4, strategy="basic")
    assert m._dot_perm is not None
    os.environ["DE_PACKED"] = "0"
    try:
        with torch.device("cuda"):
            m2 = DLRM(sizes, embedding_dim=128, bottom_mlp_dims=(64, 128),
                      top_mlp_dims=(64, 1), num_numerical=4, strategy="basic")
    finally:
        del os.environ["DE_PACKED"]
    m2.load_state_dict(m.state_dict())
    num = torch.rand(64, 4, device="cuda")
    cats = [torch.randint(0, s, (64,), device="cuda") for s in sizes]
    with torch.autocast("cuda", dtype=torch.bfloat16):
        o1 = m(num, cats)
        o2 = m2(num, cats)
    err = float((o1.float() - o2.float()).abs().max())
    assert err < 0.05, err
    o1.float().square().sum().backward()


@requires_gpu
@pytest.mark.parametrize("width", [16, 64, 96, 128, 256])
def test_csr_forward_bf16_out(width):
    """out_bf16=True stores bf16 directly — equals fp32 result RNE-rounded."""
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()
    torch.manual_seed(width)
    vocab = 700
    w = torch.randn(vocab, width, device="cuda")
    ids = torch.randint(0, vocab, (3000,), device="cuda")
    splits = torch.arange(0, 3001, 3, device="cuda")
    out32 = ext.csr_lookup_forward(w, ids, splits, False)
    out16 = ext.csr_lookup_forward(w, ids, splits, False, True)
    assert out16.dtype == torch.bfloat16
    assert torch.equal(out16, out32.bfloat16())


@requires_gpu
def test_csr_backward_bf16_grad():
    """The sparse backward consumes bf16 upstream grads natively."""
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()
    torch.manual_seed(5)
    vocab, width = 500, 64
    ids = torch.randint(0, vocab, (4000,), device="cuda")
    splits = torch.arange(0, 4001, 4, device="cuda")
    g32 = torch.randn(1000, width, device="cuda")
    g16 = g32.bfloat16()
    u32, gr32 = ext.csr_lookup_backward(g16.float(), ids, splits, vocab, False)
    u16, gr16 = ext.csr_lookup_backward(g16, ids, splits, vocab, False)
    assert torch.equal(u32, u16)
    assert gr16.dtype == torch.float32
    assert torch.allclose(gr32, gr16, atol=1e-5), \
        float((gr32 - gr16).abs().max())


@requires_gpu
def test_fused_sgd_bf16_grad_and_out():
    """Fused in-backward SGD with bf16 lookup output: the update equals the
    explicit fp32-grad update up to bf16 grad rounding."""
    from distributed_embeddings_amd.ops.embedding_lookup import (
        Ragged, csr_lookup_fused_sgd)
    torch.manual_seed(7)
    vocab, width = 300, 128
    w0 = torch.randn(vocab, width, device="cuda")
    ids = torch.randint(0, vocab, (2000,), device="cuda")
    splits = torch.arange(0, 2001, 2, device="cuda")
    lr = torch.tensor([0.5], device="cuda")

    w_a = w0.clone().requires_grad_(True)
    out = csr_lookup_fused_sgd(w_a, ids, splits, "sum", lr,
                               out_dtype=torch.bfloat16)
    assert out.dtype == torch.bfloat16
    gout = torch.randn_like(out)  # bf16 upstream grad
    out.backward(gout)

    # oracle: same bf16 grads, fp32 math on CPU
    w_b = w0.cpu().clone().requires_grad_(True)
    out_b = torch.zeros(1000, width)
    idc = ids.cpu()
    for r in range(1000):
        seg = idc[2 * r:2 * r + 2]
        out_b[r] = w_b.data[seg].sum(0)
    g = gout.float().cpu()
    grad = torch.zeros(vocab, width)
    for r in range(1000):
        for k in idc[2 * r:2 * r + 2]:
            grad[k] += g[r]
    expect = w0.cpu() - 0.5 * grad
    err = (w_a.detach().cpu() - expect).abs().max()
    assert float(err) < 1e-4, float(err)


@requires_gpu
def test_dot_interact_packed_sample_major():
    """Sample-major [B, P, D] packed layout (world==1 zero-copy) == oracle."""
    from distributed_embeddings_amd.ops.dot_interact import (
        _torch_dot_interact, dot_interact_packed)
    torch.manual_seed(4)
    B, P, D = 256, 26, 128
    perm = torch.tensor(torch.randperm(P).tolist(), dtype=torch.int32,
                        device="cuda")
    packed = torch.randn(B, P, D, device="cuda").bfloat16().requires_grad_(True)
    bottom = torch.randn(B, D, device="cuda").bfloat16().requires_grad_(True)
    out = dot_interact_packed(packed, bottom, perm, pad_to=512,
                              sample_major=True)
    feats = torch.cat([bottom.detach().unsqueeze(1),
                       packed.detach().index_select(1, perm.long())], dim=1)
    ref = _torch_dot_interact(feats.float().contiguous(), 512)
    err = (out.float() - ref).abs().max() / ref.abs().max()
    assert float(err) < 0.02, float(err)
    gout = torch.randn_like(out)
    out.backward(gout)
    f2 = feats.clone().requires_grad_(True)
    _torch_dot_interact(f2, 512).backward(gout)
    errb = (bottom.grad.float() - f2.grad[:, 0, :].float()).abs().max()
    assert float(errb) < 0.5, float(errb)
    inv = torch.empty(P, dtype=torch.long)
    inv[perm.long().cpu()] = torch.arange(P)
    gp_ref = f2.grad[:, 1:, :].index_select(1, inv.cuda())
    errp = (packed.grad.float() - gp_ref.float()).abs().max()
    assert float(errp) < 0.5, float(errp)


@requires_gpu
def test_kernel_determinism_short_segments():
    """Same inputs -> bitwise-identical results, run to run: the short-segment
    forward, the sorted sparse backward and the fused SGD update are
    atomics-free (only power-law mega-segments take the atomic-combine long
    path, which is float-order nondeterministic by design)."""
    from distributed_embeddings_amd.ops import _backend
    from distributed_embeddings_amd.ops.embedding_lookup import csr_lookup_fused_sgd
    ext = _backend.ops()
    torch.manual_seed(11)
    vocab, width, rows, hot = 2_000_000, 128, 50_000, 4
    w0 = torch.randn(vocab, width, device="cuda")
    ids = torch.randint(0, vocab, (rows * hot,), device="cuda")
    splits = torch.arange(0, rows * hot + 1, hot, device="cuda")
    g = torch.randn(rows, width, device="cuda")

    o1 = ext.csr_lookup_forward(w0, ids, splits, False)
    o2 = ext.csr_lookup_forward(w0, ids, splits, False)
    assert torch.equal(o1, o2)

    u1, g1 = ext.csr_lookup_backward(g, ids, splits, vocab, False)
    u2, g2 = ext.csr_lookup_backward(g, ids, splits, vocab, False)
    assert torch.equal(u1, u2) and torch.equal(g1, g2)

    lr = torch.tensor([0.1], device="cuda")
    wa = w0.clone().requires_grad_(True)
    wb = w0.clone().requires_grad_(True)
    for wx in (wa, wb):
        from distributed_embeddings_amd.ops.embedding_lookup import Ragged
        out = csr_lookup_fused_sgd(wx, ids, splits, "sum", lr)
        out.backward(g)
    assert torch.equal(wa.detach(), wb.detach())


@requires_gpu
@pytest.mark.parametrize("mean", [False, True])
def test_fused_adagrad_bf16_grad(mean):
    """Fused Adagrad with bf16 upstream grads (scratch+finalize GT path)
    equals the fp32-grad update."""
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()
    torch.manual_seed(9)
    vocab, width = 400, 64
    ids = torch.randint(0, vocab, (3000,), device="cuda")
    splits = torch.arange(0, 3001, 3, device="cuda")
    g16 = torch.randn(1000, width, device="cuda").bfloat16()
    lr = torch.tensor([0.3], device="cuda")

    w_a = torch.randn(vocab, width, device="cuda")
    st_a = torch.zeros_like(w_a)
    w_b, st_b = w_a.clone(), st_a.clone()
    ext.csr_fused_optimizer_apply(w_a, st_a, ids, splits, g16, lr,
                                  mean, True, 1e-10)
    ext.csr_fused_optimizer_apply(w_b, st_b, ids, splits, g16.float(), lr,
                                  mean, True, 1e-10)
    torch.cuda.synchronize()
    assert torch.allclose(w_a, w_b, atol=1e-5), \
        float((w_a - w_b).abs().max())
    assert torch.allclose(st_a, st_b, atol=1e-4)


@requires_gpu
def test_csr_backward_bf16_grad_mean():
    """bf16 grads through the MEAN-combiner backward (per-id weights + GT)."""
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()
    torch.manual_seed(12)
    vocab, width = 300, 96
    ids = torch.randint(0, vocab, (5000,), device="cuda")
    splits = torch.arange(0, 5001, 5, device="cuda")
    g16 = torch.randn(1000, width, device="cuda").bfloat16()
    u1, g1 = ext.csr_lookup_backward(g16, ids, splits, vocab, True)
    u2, g2 = ext.csr_lookup_backward(g16.float(), ids, splits, vocab, True)
    assert torch.equal(u1, u2)
    assert torch.allclose(g1, g2, atol=1e-5), float((g1 - g2).abs().max())


@requires_gpu
def test_csr_forward_bf16_out_mega_segment():
    """bf16-out disables the long-segment split (fp32 atomics unavailable);
    a power-law mega-row must still reduce correctly in-register."""
    from distributed_embeddings_amd.ops import _backend
    ext = _backend.ops()
    torch.manual_seed(3)
    vocab, width = 100, 128
    w = torch.randn(vocab, width, device="cuda")
    # row 0: 50k ids (mega-segment), rows 1..64: hotness 3
    big = torch.randint(0, vocab, (50_000,), device="cuda")
    rest = torch.randint(0, vocab, (64 * 3,), device="cuda")
    ids = torch.cat([big, rest])
    splits = torch.cat([torch.tensor([0], device="cuda"),
                        torch.tensor([50_000], device="cuda"),
                        50_000 + 3 * torch.arange(1, 65, device="cuda")])
    out16 = ext.csr_lookup_forward(w, ids, splits, True, True)  # mean, bf16
    out32 = ext.csr_lookup_forward(w, ids, splits, True)
    assert out16.dtype == torch.bfloat16
    err = (out16.float() - out32).abs().max()
    assert float(err) < 0.02, float(err)


@requires_gpu
def test_inference_graph_capture_matches_eager():
    """Serving path: hipGraph-captured eval scoring == eager scoring."""
    import distributed_embeddings_amd as de
    from distributed_embeddings_amd.models.dlrm import DLRM
    torch.manual_seed(0)
    sizes = [500, 600, 700]
    with torch.device("cuda"):
        m = DLRM(sizes, embedding_dim=128, bottom_mlp_dims=(32, 128),
                 top_mlp_dims=(32, 1), num_numerical=4)
    m.eval()
    num = torch.rand(256, 4, device="cuda")
    cats = [torch.randint(0, s, (256,), device="cuda") for s in sizes]

    def score():
        with torch.autocast("cuda", dtype=torch.bfloat16):
            return torch.sigmoid(m(num, cats).float())

    with torch.inference_mode():
        eager = score().clone()
        for _ in range(3):
            out = score()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            out = score()
        g.replay()
        torch.cuda.synchronize()
    assert torch.allclose(out, eager, atol=1e-3), \
        float((out - eager).abs().max())
