"""Rank-group math (no torch.distributed needed)."""

import pytest

from megatron_amd.parallel.grid import ParallelGrid, compose, decompose, orthogonal_rank_groups
from megatron_amd.parallel import grid as G


def test_decompose_compose_roundtrip():
    shape = [2, 3, 4]
    for i in range(24):
        assert compose(decompose(i, shape), shape) == i


def test_orthogonal_groups_tp_dp():
    # world 8 = tp2 * cp1 * dp2 * pp2
    shape = [2, 1, 2, 2]
    tp_groups = orthogonal_rank_groups(8, shape, [True, False, False, False])
    assert sorted(map(tuple, tp_groups)) == [(0, 1), (2, 3), (4, 5), (6, 7)]
    dp_groups = orthogonal_rank_groups(8, shape, [False, False, True, False])
    assert sorted(map(tuple, dp_groups)) == [(0, 2), (1, 3), (4, 6), (5, 7)]
    pp_groups = orthogonal_rank_groups(8, shape, [False, False, False, True])
    assert sorted(map(tuple, pp_groups)) == [(0, 4), (1, 5), (2, 6), (3, 7)]


def test_every_rank_in_exactly_one_group():
    shape = [2, 2, 2, 2]
    for mask in ([True, False, True, False], [False, True, False, True], [True, True, False, False]):
        groups = orthogonal_rank_groups(16, shape, mask)
        seen = sorted(r for g in groups for r in g)
        assert seen == list(range(16))


@pytest.mark.parametrize("tp,cp,pp,world", [(2, 1, 2, 8), (4, 1, 2, 8), (2, 2, 2, 8), (1, 1, 1, 1)])
def test_grid_shapes(tp, cp, pp, world):
    for rank in range(world):
        g = ParallelGrid(
            tensor_parallel_size=tp, pipeline_parallel_size=pp, context_parallel_size=cp,
            world_size=world, rank=rank,
        )
        assert g.tp * g.cp * g.dp * g.pp == world
        assert g.rank in g.ranks("tp")
        assert g.rank in g.ranks("dp")
        assert g.rank in g.ranks("pp")
        assert len(g.ranks("tp")) == tp
        assert len(g.ranks("pp")) == pp
        # coords consistency
        assert g.ranks("tp")[g.tp_rank] == g.rank
        assert g.ranks("pp")[g.pp_rank] == g.rank


def test_embedding_group_first_last_stage():
    g = ParallelGrid(tensor_parallel_size=2, pipeline_parallel_size=2, world_size=8, rank=0)
    # embd group for rank 0: first and last pp stage of its pipeline
    assert g.ranks("embd") == [0, 4]


def test_expert_groups():
    # world 8: tp2 dp4; ep2 -> etp2, ep2, edp2
    g = ParallelGrid(tensor_parallel_size=2, expert_parallel_size=2, world_size=8, rank=0)
    assert g.ep == 2 and g.edp == 2
    assert len(g.ranks("ep")) == 2
    assert len(g.ranks("expert_dp")) == 2
    seen = set(g.ranks("ep")) | set(g.ranks("expert_dp")) | set(g.ranks("etp"))
    assert 0 in seen


def test_hyper_comm_grid_coords():
    from megatron_amd.parallel.hyper_grid import HyperCommGrid

    g = HyperCommGrid(["tp", "dp", "pp"], [2, 3, 2])
    assert g.total == 12
    # round trip
    for r in range(12):
        assert g.rank_at(**g.coords_of(r)) == r
    # tp fastest-varying
    assert g.coords_of(1) == {"tp": 1, "dp": 0, "pp": 0}
    assert g.coords_of(2) == {"tp": 0, "dp": 1, "pp": 0}
    assert g.coords_of(6) == {"tp": 0, "dp": 0, "pp": 1}
    # group spans
    assert g.ranks_for(["tp"], 3) == [2, 3]
    assert g.ranks_for(["dp"], 1) == [1, 3, 5]
    assert g.ranks_for(["tp", "dp"], 7) == [6, 7, 8, 9, 10, 11]
    groups = g.all_groups_for(["dp"])
    assert len(groups) == 4  # 2 tp x 2 pp
    flat = sorted(r for grp in groups for r in grp)
    assert flat == list(range(12))


def test_hyper_comm_grid_offset_subgrid():
    from megatron_amd.parallel.hyper_grid import HyperCommGrid

    # a vision-encoder sub-grid occupying ranks 8..11 next to an LLM grid
    g = HyperCommGrid(["tp", "dp"], [2, 2], rank_offset=8)
    assert g.ranks_for(["tp"], 10) == [10, 11]
    assert g.coords_of(9) == {"tp": 1, "dp": 0}


def _hyper_groups_case(rank, world):
    import torch
    import torch.distributed as dist

    from megatron_amd.parallel.hyper_grid import HyperCommGrid

    G.initialize_model_parallel()
    g = HyperCommGrid(["a", "b"], [2, 1])
    grp = g.group_for(["a"])
    assert dist.get_world_size(grp) == 2
    t = torch.tensor([float(rank + 1)])
    dist.all_reduce(t, group=grp)
    assert float(t) == 3.0


def test_hyper_comm_grid_groups_gloo():
    from tests.utils import spawn_dist

    spawn_dist(_hyper_groups_case, 2)


def test_comm_config_options(tmp_path):
    from megatron_amd.parallel.comm_config import comm_config, load_comm_config, pg_options_for

    cfg = tmp_path / "nccl.yaml"
    cfg.write_text("tp:\n  max_ctas: 8\n  is_high_priority_stream: true\ndp:\n  min_ctas: 2\n")
    load_comm_config(str(cfg))
    assert comm_config()["tp"]["max_ctas"] == 8
    # CPU container: options resolve to None (gloo ignores them) but the
    # config is validated
    assert pg_options_for("tp") is None or pg_options_for("tp").is_high_priority_stream
    assert pg_options_for("pp") is None  # unconfigured group
    import pytest as _pytest

    with _pytest.raises(ValueError, match="unknown comm-config"):
        load_comm_config({"tp": {"bogus_knob": 1}})
    load_comm_config({})  # reset
