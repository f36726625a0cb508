"""End-to-end DLRM distributed equivalence (gloo world=2) — exercises the
exact bench.py path: hybrid dp+mp, fused group lookups, output_dtype-cast
all-to-all, DistributedOptimizer, fused SGD."""

import numpy as np
import pytest
import torch

from conftest import run_distributed

SIZES = [50, 7, 120, 33, 64, 200]


def _dlrm_worker(rank, world, fused_sgd, output_dtype_name):
    import distributed_embeddings_amd as de
    from distributed_embeddings_amd.models.dlrm import DLRM
    from distributed_embeddings_amd.parallel.optim import SparseEmbeddingOptimizer

    torch.manual_seed(0)
    model = DLRM(SIZES, embedding_dim=16, bottom_mlp_dims=(32, 16),
                 top_mlp_dims=(32, 1), num_numerical=4,
                 strategy="memory_balanced")
    # deterministic weights everywhere
    gw = torch.Generator().manual_seed(42)
    weights = [torch.randn(s, 16, generator=gw).numpy() for s in SIZES]
    model.embeddings.set_weights(weights)
    for p in model.bottom_mlp.parameters():
        torch.nn.init.normal_(p, generator=gw) if p.dim() > 1 else p.data.zero_()
    for p in model.top_mlp.parameters():
        torch.nn.init.normal_(p, generator=gw) if p.dim() > 1 else p.data.zero_()
    de.broadcast_parameters(model)

    if fused_sgd:
        model.embeddings.enable_fused_sgd(0.05)
    opt = de.DistributedOptimizer(
        SparseEmbeddingOptimizer(model.parameters(), lr=0.05), average=False)

    gi = torch.Generator().manual_seed(7)
    B = 8
    cats = [torch.randint(0, s, (B,), generator=gi) for s in SIZES]
    num = torch.rand(B, 4, generator=gi)
    labels = torch.randint(0, 2, (B, 1), generator=gi).float()
    lb = B // world
    sl = slice(rank * lb, (rank + 1) * lb)

    for _ in range(2):
        opt.zero_grad(set_to_none=True)
        logits = model(num[sl], [c[sl] for c in cats])
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            logits.float(), labels[sl], reduction="sum") / B
        loss.backward()
        opt.step()
    out = model(num[sl], [c[sl] for c in cats]).detach()
    tables = model.embeddings.get_weights(all_ranks=True)
    return {"out": out, "tables": [torch.as_tensor(t) for t in tables]}


@pytest.mark.parametrize("fused_sgd", [False, True])
def test_dlrm_world2_matches_world1(fused_sgd):
    results2 = run_distributed(_dlrm_worker, world=2, args=(fused_sgd, None))
    results1 = run_distributed(_dlrm_worker, world=1, args=(fused_sgd, None))
    full_out = results1[0]["out"]
    for rank in range(2):
        got = results2[rank]["out"]
        ref = full_out[rank * 4:(rank + 1) * 4]
        assert torch.allclose(got, ref, atol=1e-4), \
            f"rank{rank} fwd err {(got - ref).abs().max()}"
    for t in range(len(SIZES)):
        a = results2[0]["tables"][t]
        b = results1[0]["tables"][t]
        assert torch.allclose(a, b, atol=1e-4), \
            f"table {t} err {(a - b).abs().max()}"


def _dtype_worker(rank, world):
    import distributed_embeddings_amd as de
    tables = [de.TableConfig(40, 8), de.TableConfig(60, 8)]
    model = de.DistributedEmbedding(tables, strategy="basic")
    gw = torch.Generator().manual_seed(1)
    weights = [torch.randn(40, 8, generator=gw).numpy(),
               torch.randn(60, 8, generator=gw).numpy()]
    model.set_weights(weights)
    gi = torch.Generator().manual_seed(2)
    inputs = [torch.randint(0, 40, (world * 4,), generator=gi),
              torch.randint(0, 60, (world * 4,), generator=gi)]
    sl = slice(rank * 4, (rank + 1) * 4)
    outs = model([x[sl] for x in inputs], output_dtype=torch.bfloat16)
    refs = [torch.from_numpy(weights[t])[inputs[t][sl]].bfloat16()
            for t in range(2)]
    assert all(o.dtype == torch.bfloat16 for o in outs)
    return [float((o.float() - r.float()).abs().max()) for o, r in zip(outs, refs)]


def test_output_dtype_bf16_a2a_world2():
    results = run_distributed(_dtype_worker, world=2)
    for errs in results:
        assert max(errs) < 0.05  # bf16 rounding only


def _bcast_callback_worker(rank, world):
    import distributed_embeddings_amd as de
    m = torch.nn.Linear(4, 4)
    with torch.no_grad():
        m.weight.fill_(float(rank))
    cb = de.BroadcastParametersOnFirstStep(m)
    cb()  # first step: broadcast from rank 0
    first = m.weight.detach().clone()
    with torch.no_grad():
        m.weight.fill_(float(rank) + 10)
    cb()  # later steps: no-op
    return {"first": first, "after": m.weight.detach().clone()}


def test_broadcast_on_first_step_world2():
    results = run_distributed(_bcast_callback_worker, world=2)
    for r in range(2):
        assert torch.equal(results[r]["first"], torch.zeros(4, 4))
        assert float(results[r]["after"][0, 0]) == r + 10


def test_dlrm_world4_matches_world1():
    """World-4 hybrid dp+mp DLRM == single process (the CPU proxy for the
    driver's 8-GPU scaling bench; B=8 -> 2 samples/rank)."""
    results4 = run_distributed(_dlrm_worker, world=4, args=(True, None))
    results1 = run_distributed(_dlrm_worker, world=1, args=(True, None))
    full_out = results1[0]["out"]
    for rank in range(4):
        got = results4[rank]["out"]
        ref = full_out[rank * 2:(rank + 1) * 2]
        assert torch.allclose(got, ref, atol=1e-4), \
            f"rank{rank} fwd err {(got - ref).abs().max()}"
    for t in range(len(SIZES)):
        a = results4[0]["tables"][t]
        b = results1[0]["tables"][t]
        assert torch.allclose(a, b, atol=1e-4), \
            f"table {t} err {(a - b).abs().max()}"


def test_dlrm_world8_matches_world1():
    """World-8 (the driver's max scaling point): one sample per rank."""
    results8 = run_distributed(_dlrm_worker, world=8, args=(True, None),
                               timeout=300)
    results1 = run_distributed(_dlrm_worker, world=1, args=(True, None))
    full_out = results1[0]["out"]
    for rank in range(8):
        got = results8[rank]["out"]
        ref = full_out[rank:rank + 1]
        assert torch.allclose(got, ref, atol=1e-4), \
            f"rank{rank} fwd err {(got - ref).abs().max()}"
    for t in range(len(SIZES)):
        a = results8[0]["tables"][t]
        b = results1[0]["tables"][t]
        assert torch.allclose(a, b, atol=1e-4), \
            f"table {t} err {(a - b).abs().max()}"


def _synth_worker(rank, world):
    import distributed_embeddings_amd as de
    from distributed_embeddings_amd.models.config import EmbeddingConfig, ModelConfig
    from distributed_embeddings_amd.models.synthetic import SyntheticModel
    from distributed_embeddings_amd.utils.input_gen import make_batch
    from distributed_embeddings_amd.models.synthetic import expand_tables

    cfg = ModelConfig(
        name="unit",
        embedding_configs=[
            EmbeddingConfig(2, [1, 3], 50, 8, True),   # shared multi-hot
            EmbeddingConfig(3, [1], 80, 8, False),
        ],
        mlp_sizes=[16, 8], num_numerical_features=4, interact_stride=2)
    torch.manual_seed(11)
    model = SyntheticModel(cfg, strategy="memory_balanced")
    tables, input_map, hotness = expand_tables(cfg)
    gw = torch.Generator().manual_seed(9)
    weights = [torch.randn(r, w, generator=gw).numpy() for r, w in tables]
    model.embeddings.set_weights(weights)
    for p in model.mlp.parameters():
        torch.nn.init.normal_(p, generator=gw) if p.dim() > 1 else p.data.zero_()
    de.broadcast_parameters(model)

    B = 8
    gi = torch.Generator().manual_seed(13)
    sizes = [tables[t][0] for t in input_map]
    cats = make_batch(sizes, hotness, B, generator=gi, keep_hot_dim=True)
    num = torch.rand(B, 4, generator=gi)
    lb = B // world
    sl = slice(rank * lb, (rank + 1) * lb)
    out = model(num[sl], [c[sl] for c in cats]).detach()
    return {"out": out}


def test_synthetic_model_world2_matches_world1():
    r2 = run_distributed(_synth_worker, world=2)
    r1 = run_distributed(_synth_worker, world=1)
    full = r1[0]["out"]
    for rank in range(2):
        ref = full[rank * 4:(rank + 1) * 4]
        got = r2[rank]["out"]
        assert torch.allclose(got, ref, atol=1e-4), \
            f"rank{rank} err {(got - ref).abs().max()}"


def _packed_vs_legacy_worker(rank, world):
    """Packed interaction path == legacy stack path, world 2 (gloo)."""
    import os
    import distributed_embeddings_amd as de
    from distributed_embeddings_amd.models.dlrm import DLRM
    sizes = [50, 60, 70, 40]
    torch.manual_seed(0)
    m = DLRM(sizes, embedding_dim=32, bottom_mlp_dims=(64, 32),
             top_mlp_dims=(64, 1), num_numerical=4, strategy="basic")
    de.broadcast_parameters(m)
    assert m._dot_perm is not None
    os.environ["DE_PACKED"] = "0"
    try:
        m2 = DLRM(sizes, embedding_dim=32, bottom_mlp_dims=(64, 32),
                  top_mlp_dims=(64, 1), num_numerical=4, strategy="basic")
    finally:
        del os.environ["DE_PACKED"]
    assert m2._dot_perm is None
    m2.load_state_dict(m.state_dict())
    m2.embeddings.set_weights(m.embeddings.get_weights(all_ranks=True))

    g = torch.Generator().manual_seed(7)
    B = world * 4
    num = torch.rand(B, 4, generator=g)
    cats = [torch.randint(0, s, (B,), generator=g) for s in sizes]
    sl = slice(rank * 4, (rank + 1) * 4)
    o1 = m(num[sl], [c[sl] for c in cats])
    o2 = m2(num[sl], [c[sl] for c in cats])
    err = float((o1 - o2).abs().max())
    # grads flow through the packed view + a2a
    o1.square().sum().backward()
    return err


def test_dlrm_packed_matches_legacy_world2():
    errs = run_distributed(_packed_vs_legacy_worker, world=2)
    assert max(errs) < 1e-5, errs


def test_dlrm_packed_matches_legacy_world1():
    err = _packed_vs_legacy_worker(0, 1)
    assert err < 1e-6, err
