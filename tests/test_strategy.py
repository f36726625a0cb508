"""Planner unit tests (pure host, no GPU).

Covers the reference planner behaviors: grouping thresholds,
the three placement strategies, column slicing + merge, auto-threshold with
fewer tables than workers, row slicing with remainders, offload marking,
concat groups, shared tables.
"""

import pytest

from distributed_embeddings_amd import DistEmbeddingStrategy, TableConfig


def make(sizes, width=8, combiner=None):
    return [TableConfig(s, width, combiner) for s in sizes]


def test_grouping_thresholds():
    plan = DistEmbeddingStrategy(
        make([10, 1000, 100000]), 2,
        data_parallel_threshold=10 * 8,
        row_slice_threshold=100000 * 8,
    )
    dp, col, row = plan.table_groups
    assert dp == [0] and col == [1] and row == [2]


def test_basic_round_robin():
    plan = DistEmbeddingStrategy(make([100, 200, 300, 400]), 2, strategy="basic")
    t_per_rank = [[s.table_id for s in plan.rank_slices[r]] for r in range(2)]
    assert t_per_rank == [[0, 2], [1, 3]]


def test_memory_balanced_counts():
    sizes = [100, 900, 300, 700, 500, 600]
    plan = DistEmbeddingStrategy(make(sizes), 2, strategy="memory_balanced")
    per_rank = [plan.rank_slices[r] for r in range(2)]
    # equal counts, roughly equal memory
    assert len(per_rank[0]) == len(per_rank[1]) == 3
    mem = [sum(s.width * sizes[s.table_id] // 8 for s in slices) for slices in per_rank]
    assert abs(mem[0] - mem[1]) <= max(sizes)


def test_memory_optimized_balance():
    sizes = [1000, 10, 10, 10, 10, 10]
    plan = DistEmbeddingStrategy(make(sizes), 2, strategy="memory_optimized")
    mems = []
    for r in range(2):
        mems.append(sum(8 * sizes[s.table_id] for s in plan.rank_slices[r]))
    # big table alone on one rank, all small ones on the other
    assert min(len(plan.rank_slices[0]), len(plan.rank_slices[1])) == 1


def test_column_slice_threshold():
    # one 64-wide table of 1000 rows = 64000 elems; threshold 20000 -> 4 slices
    plan = DistEmbeddingStrategy([TableConfig(1000, 64)], 4,
                                 column_slice_threshold=20000)
    slices = plan.table_slices[0]
    assert len(slices) == 4
    assert [s.width for s in slices] == [16, 16, 16, 16]
    assert [s.col_offset for s in slices] == [0, 16, 32, 48]
    assert sorted(s.rank for s in slices) == [0, 1, 2, 3]


def test_column_slice_merge_same_rank():
    # 4 slices, 2 ranks -> 2 slices per rank, merged into one wider slice each
    plan = DistEmbeddingStrategy([TableConfig(1000, 64)], 2,
                                 column_slice_threshold=20000)
    slices = plan.table_slices[0]
    assert len(slices) == 2
    assert [s.width for s in slices] == [32, 32]
    assert {s.rank for s in slices} == {0, 1}


def test_auto_column_slice_fewer_tables_than_workers():
    plan = DistEmbeddingStrategy(make([1000, 2000]), 4)
    # enough slices that every rank holds something
    total_slices = sum(len(v) for v in plan.table_slices.values())
    assert total_slices >= 4
    for r in range(4):
        assert plan.rank_slices[r], f"rank {r} got no slice"


def test_row_slice_remainder():
    plan = DistEmbeddingStrategy([TableConfig(10, 4)], 3,
                                 row_slice_threshold=1)
    shards = plan.row_shards[0]
    assert [s.rows for s in shards] == [4, 3, 3]
    assert [s.row_offset for s in shards] == [0, 4, 7]


def test_offload_marks_largest():
    plan = DistEmbeddingStrategy(make([10, 1000, 100]), 1,
                                 gpu_embedding_size=(10 + 100) * 8)
    offloaded = {s.table_id: s._offload for s in plan.rank_slices[0]}
    assert offloaded == {0: False, 2: False, 1: True}


def test_concat_groups_same_width_and_combiner():
    cfgs = [TableConfig(10, 8, "sum"), TableConfig(20, 8, "sum"),
            TableConfig(30, 16, "sum"), TableConfig(40, 8, "mean")]
    plan = DistEmbeddingStrategy(cfgs, 1)
    groups = plan.local_concat_groups(0)
    assert len(groups) == 3
    fused = groups[0]
    assert fused.input_dim == 30
    assert [m.concat_row_offset for m in fused.members] == [0, 10]


def test_shared_tables_input_map():
    plan = DistEmbeddingStrategy(make([100, 200]), 2,
                                 input_table_map=[0, 1, 0])
    # input 0 and 2 share table 0
    assert plan.input_table_map == [0, 1, 0]
    # each rank serving table 0 sees both inputs
    for r in range(2):
        ids = plan.rank_input_ids[r]
        tables = [s.table_id for s in plan.rank_input_slices[r]]
        for inp, t in zip(ids, tables):
            assert plan.input_table_map[plan.input_groups[1][inp]] == t


def test_reverse_input_order():
    cfgs = make([10, 100000, 10, 100000])
    plan = DistEmbeddingStrategy(cfgs, 2, data_parallel_threshold=100)
    # groups: dp=[0,2] col=[1,3]; flat order = [0,2,1,3]; reverse restores 0..3
    flat = [i for grp in plan.input_groups for i in grp]
    restored = [flat[i] for i in plan.reverse_input_order]
    assert restored == [0, 1, 2, 3]


def test_sliced_out_ranges():
    cfgs = [TableConfig(100, 8), TableConfig(1000, 64)]
    plan = DistEmbeddingStrategy(cfgs, 2, column_slice_threshold=20000)
    nslices = len(plan.table_slices[1])
    assert nslices == 2
    assert plan.sliced_out_ranges == [(1, 1 + nslices)]


def test_invalid_strategy():
    with pytest.raises(ValueError):
        DistEmbeddingStrategy(make([10]), 1, strategy="bogus")


def test_world1_passthrough():
    plan = DistEmbeddingStrategy(make([10, 20]), 1)
    assert len(plan.rank_slices[0]) == 2


def test_planner_handles_jumbo_and_colossal_scales():
    """BASELINE scale table (tiny 4.2 GiB ... colossal 22.3 TiB): host-only
    plans on an 8-GPU 288 GB/GPU node.  'large' (774 GiB) fits entirely in
    HBM via row+table sharding; jumbo (3.1 TiB) and colossal (22.3 TiB)
    exceed aggregate HBM and must land within budget via CPU offload of the
    overflow (parity: reference offload is a TP-group feature; row-sliced
    tables are never offloaded)."""
    from distributed_embeddings_amd.models.config import synthetic_models
    from distributed_embeddings_amd.models.synthetic import expand_tables
    from distributed_embeddings_amd.parallel.strategy import (
        DistEmbeddingStrategy, TableConfig)

    budget = int(0.9 * 288e9) // 4  # fp32 elements per GPU

    def resident_elems(plan, r):
        col = sum(plan.configs[s.table_id].input_dim * s.width
                  for s in plan.rank_slices[r] if not s._offload)
        row = sum(plan.row_shards[t][r].rows * plan.configs[t].output_dim
                  for t in plan.row_table_ids)
        return col + row

    # large: fits HBM outright (row-slice the >=2e9-element tables)
    tables, imap, _ = expand_tables(synthetic_models["large"])
    cfgs = [TableConfig(r, w, "sum") for r, w in tables]
    plan = DistEmbeddingStrategy(
        cfgs, 8, strategy="memory_optimized", input_table_map=imap,
        row_slice_threshold=2_000_000_000, gpu_embedding_size=budget)
    for r in range(8):
        assert resident_elems(plan, r) <= budget
        assert not any(s._offload for s in plan.rank_slices[r])

    # jumbo / colossal: beyond aggregate HBM -> overflow offloads to CPU
    for name in ("jumbo", "colossal"):
        tables, imap, _ = expand_tables(synthetic_models[name])
        cfgs = [TableConfig(r, w, "sum") for r, w in tables]
        total = sum(r * w for r, w in tables)
        assert total > 8 * budget  # the scale genuinely demands offload
        plan = DistEmbeddingStrategy(
            cfgs, 8, strategy="memory_optimized", input_table_map=imap,
            gpu_embedding_size=budget)
        offloaded = 0
        for r in range(8):
            assert resident_elems(plan, r) <= budget, (name, r)
            offloaded += sum(plan.configs[s.table_id].input_dim * s.width
                             for s in plan.rank_slices[r] if s._offload)
        assert offloaded >= total - 8 * budget
        # every table is placed exactly once
        assert all(plan.table_slices[t] for t in plan.col_table_ids)
