"""Tests for utils: LR schedule, input generation, Criteo binary reader."""

import numpy as np
import torch

from distributed_embeddings_amd.utils.criteo import RawBinaryDataset, SyntheticDLRMData
from distributed_embeddings_amd.utils.input_gen import make_batch, power_law_ids
from distributed_embeddings_amd.utils.lr_schedule import WarmupPolyDecay


def test_warmup_poly_decay_shape():
    class Opt:
        param_groups = [{"lr": 0.0}]
    sched = WarmupPolyDecay(Opt(), base_lr=1.0, warmup_steps=10,
                            decay_start=20, decay_steps=10, power=2.0)
    lrs = [sched.lr_at(s) for s in (0, 4, 9, 10, 15, 19, 20, 25, 30, 40)]
    assert abs(lrs[0] - 0.1) < 1e-9          # warmup start
    assert abs(lrs[2] - 1.0) < 1e-9          # warmup end
    assert lrs[3] == lrs[4] == lrs[5] == 1.0  # plateau
    assert 0 < lrs[7] < 1.0                  # decaying
    assert lrs[8] == 0.0 and lrs[9] == 0.0   # decayed

    # step() writes into param_groups
    o = Opt()
    s = WarmupPolyDecay(o, base_lr=2.0, warmup_steps=2)
    s.step()
    assert abs(o.param_groups[0]["lr"] - 1.0) < 1e-9


def test_power_law_ids_skew():
    torch.manual_seed(0)
    ids = power_law_ids(10000, (200000,), alpha=1.05)
    assert int(ids.min()) >= 0 and int(ids.max()) < 10000
    # head ids much more frequent than tail
    head = (ids < 10).float().mean()
    tail = ((ids >= 9000)).float().mean()
    assert head > 10 * tail


def test_make_batch_shapes():
    cats = make_batch([100, 200], [1, 4], 8)
    assert cats[0].shape == (8,)
    assert cats[1].shape == (8, 4)
    cats = make_batch([100], [1], 8, keep_hot_dim=True)
    assert cats[0].shape == (8, 1)


def test_raw_binary_dataset_round_trip(tmp_path):
    # build a tiny split-binary dataset (reference layout)
    n, nnum = 32, 3
    sizes = [50, 70000]
    d = tmp_path / "train"
    d.mkdir()
    rng = np.random.RandomState(0)
    labels = rng.randint(0, 2, n).astype(np.int8)
    nums = rng.rand(n, nnum).astype(np.float16)
    cats = [rng.randint(0, s, n) for s in sizes]
    (d / "label.bin").write_bytes(labels.tobytes())
    (d / "numerical.bin").write_bytes(nums.tobytes())
    (d / "cat_0.bin").write_bytes(cats[0].astype(np.int8).tobytes())
    (d / "cat_1.bin").write_bytes(cats[1].astype(np.int32).tobytes())

    ds = RawBinaryDataset(str(tmp_path), batch_size=8,
                          categorical_features=[0, 1],
                          categorical_feature_sizes=sizes,
                          num_numerical=nnum, device="cpu")
    assert len(ds) == 4
    batches = list(ds)
    assert len(batches) == 4
    num, cat_list, lab = batches[0]
    assert num.shape == (8, nnum) and lab.shape == (8, 1)
    assert torch.equal(cat_list[0], torch.from_numpy(cats[0][:8].astype(np.int64)))
    assert torch.equal(cat_list[1], torch.from_numpy(cats[1][:8].astype(np.int64)))
    assert np.allclose(num.numpy(), nums[:8].astype(np.float32))


def test_raw_binary_dataset_mp_input(tmp_path):
    n = 16
    d = tmp_path / "train"
    d.mkdir()
    rng = np.random.RandomState(1)
    (d / "label.bin").write_bytes(rng.randint(0, 2, n).astype(np.int8).tobytes())
    (d / "numerical.bin").write_bytes(rng.rand(n, 2).astype(np.float16).tobytes())
    cat = rng.randint(0, 1000, n)
    (d / "cat_0.bin").write_bytes(cat.astype(np.int16).tobytes())
    ds = RawBinaryDataset(str(tmp_path), batch_size=8, categorical_features=[0],
                          categorical_feature_sizes=[1000], num_numerical=2,
                          rank=1, world=2, dp_input=False, device="cpu")
    num, cat_list, lab = next(iter(ds))
    # mp input: full global batch of this rank's features
    assert cat_list[0].shape == (8,)
    assert torch.equal(cat_list[0], torch.from_numpy(cat[:8].astype(np.int64)))
    # labels/numericals stay local-batch
    assert num.shape == (4, 2) and lab.shape == (4, 1)


def test_synthetic_dlrm_data():
    data = SyntheticDLRMData([100, 200], local_bs=8, num_batches=3,
                             num_numerical=4)
    batches = list(data)
    assert len(batches) == 3
    num, cats, labels = batches[0]
    assert num.shape == (8, 4) and len(cats) == 2 and labels.shape == (8, 1)


def test_warmup_poly_decay_updates_fused_lr():
    import torch
    from distributed_embeddings_amd import Embedding

    class Opt:
        param_groups = [{"lr": 0.0}]

    emb = Embedding(10, 4, combiner="sum")
    emb.enable_fused_sgd(1.0)
    sched = WarmupPolyDecay(Opt(), base_lr=1.0, warmup_steps=4,
                            fused_modules=[emb])
    sched.step()
    assert abs(float(emb._fused_lr) - 0.25) < 1e-6
    for _ in range(3):
        sched.step()
    assert abs(float(emb._fused_lr) - 1.0) < 1e-6


def test_cat_dtype_boundaries():
    """Signed-dtype boundaries (parity: reference get_categorical_feature_type
    uses iinfo(dtype).max): vocab 200 must NOT be int8 (ids >=128 would wrap
    negative)."""
    from distributed_embeddings_amd.utils.criteo import _cat_dtype
    assert _cat_dtype(100) == (np.int8, 1)
    assert _cat_dtype(126) == (np.int8, 1)
    assert _cat_dtype(127) == (np.int16, 2)
    assert _cat_dtype(200) == (np.int16, 2)
    assert _cat_dtype(32766) == (np.int16, 2)
    assert _cat_dtype(32767) == (np.int32, 4)
    assert _cat_dtype(40_000_000) == (np.int32, 4)
