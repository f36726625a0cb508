"""Property-based planner invariants (hypothesis)."""

from hypothesis import given, settings, strategies as st

from distributed_embeddings_amd import DistEmbeddingStrategy, TableConfig


@st.composite
def plan_inputs(draw):
    n_tables = draw(st.integers(1, 12))
    sizes = [draw(st.integers(1, 5000)) for _ in range(n_tables)]
    widths = [draw(st.sampled_from([1, 4, 8, 16, 32, 64, 96])) for _ in range(n_tables)]
    world = draw(st.sampled_from([1, 2, 3, 4, 8]))
    strategy = draw(st.sampled_from(["basic", "memory_balanced", "memory_optimized"]))
    col_thr = draw(st.one_of(st.none(), st.integers(1, 200000)))
    dp_thr = draw(st.one_of(st.none(), st.integers(0, 10000)))
    row_thr = draw(st.one_of(st.none(), st.integers(1000, 10 ** 6)))
    n_inputs = draw(st.integers(n_tables, n_tables + 4))
    extra = [draw(st.integers(0, n_tables - 1)) for _ in range(n_inputs - n_tables)]
    input_map = list(range(n_tables)) + extra
    return sizes, widths, world, strategy, col_thr, dp_thr, row_thr, input_map


@settings(max_examples=50, deadline=None)
@given(plan_inputs())
def test_plan_invariants(inp):
    sizes, widths, world, strategy, col_thr, dp_thr, row_thr, input_map = inp
    cfgs = [TableConfig(s, w) for s, w in zip(sizes, widths)]
    plan = DistEmbeddingStrategy(
        cfgs, world, strategy=strategy, input_table_map=input_map,
        column_slice_threshold=col_thr, data_parallel_threshold=dp_thr,
        row_slice_threshold=row_thr)

    dp, col, row = plan.table_groups
    # every table in exactly one group
    assert sorted(dp + col + row) == list(range(len(cfgs)))

    # column slices of each col table tile its width exactly, in order
    for t in col:
        slices = plan.table_slices[t]
        off = 0
        for s in slices:
            assert s.col_offset == off
            assert s.width > 0
            off += s.width
        assert off == cfgs[t].output_dim

    # row shards cover the vocab exactly
    for t in row:
        shards = plan.row_shards[t]
        assert len(shards) == world
        assert sum(s.rows for s in shards) == cfgs[t].input_dim
        off = 0
        for s in shards:
            assert s.row_offset == off
            off += s.rows

    # concat groups partition each rank's slices; row offsets consistent
    for r in range(world):
        seen = set()
        for grp in plan.rank_concat_groups[r]:
            total = 0
            for m in grp.members:
                assert m.concat_row_offset == total
                total += cfgs[m.table_id].input_dim
                assert id(m) not in seen
                seen.add(id(m))
            assert total == grp.input_dim
        assert len(seen) == len(plan.rank_slices[r])

    # reverse input order is a permutation restoring original order
    flat = [i for grp in plan.input_groups for i in grp]
    assert sorted(flat) == list(range(len(input_map)))
    restored = [flat[i] for i in plan.reverse_input_order]
    assert restored == list(range(len(input_map)))

    # worker-order pair list + rev_tp_order is consistent
    worker_inputs = [i for r in range(world) for i in plan.rank_input_ids[r]]
    reordered = [worker_inputs[i] for i in plan.rev_tp_order]
    assert reordered == sorted(worker_inputs)
