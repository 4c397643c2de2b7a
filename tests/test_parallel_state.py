"""Topology rank-list math, tested without process groups (mesh_only mode),
mirroring the reference's ground-truth-list tests
(test/unit_test/parallel_layers/test_parallel_state.py:19-60)."""

import pytest
import torch

from neuronx_distributed_amd.parallel import parallel_state as ps


def _mesh(world, tp=1, pp=1, cp=1, ep=1):
    ps._MPU_WORLD_SIZE = world
    try:
        return ps.initialize_model_parallel(
            tensor_model_parallel_size=tp,
            pipeline_model_parallel_size=pp,
            context_parallel_size=cp,
            expert_model_parallel_size=ep,
            mesh_only=True,
        )
    finally:
        ps._MPU_WORLD_SIZE = None


def test_tp_dp_8():
    g = _mesh(8, tp=2)
    assert g.tp_groups == [[0, 1], [2, 3], [4, 5], [6, 7]]
    assert g.dp_groups == [[0, 2, 4, 6], [1, 3, 5, 7]]
    assert g.pp_groups == [[0], [1], [2], [3], [4], [5], [6], [7]]


def test_tp_pp_8():
    g = _mesh(8, tp=2, pp=2)
    assert g.tp_groups == [[0, 1], [2, 3], [4, 5], [6, 7]]
    assert g.dp_groups == [[0, 2], [1, 3], [4, 6], [5, 7]]
    assert g.pp_groups == [[0, 4], [1, 5], [2, 6], [3, 7]]


def test_128_rank_pp2_cp4_tp8():
    # reference ground truth: pp2 dp2 cp4 tp8 on 128 ranks
    g = _mesh(128, tp=8, pp=2, cp=4)
    assert len(g.tp_groups) == 16 and all(len(x) == 8 for x in g.tp_groups)
    assert g.tp_groups[0] == list(range(8))
    # every rank in exactly one group of each kind
    for groups, deg in ((g.tp_groups, 8), (g.dp_groups, 2), (g.pp_groups, 2),
                        (g.cp_groups, 4)):
        seen = sorted(r for grp in groups for r in grp)
        assert seen == list(range(128))
        assert all(len(x) == deg for x in groups)
    # CP groups stride by tp within one pp/dp block
    assert g.cp_groups[0] == [0, 8, 16, 24]
    # DP groups stride by cp*tp
    assert g.dp_groups[0] == [0, 32]
    # PP outermost
    assert g.pp_groups[0] == [0, 64]


def test_expert_mesh():
    g = _mesh(16, tp=2, ep=4)
    # dp = 8, ep=4 -> dp_exp = 2
    assert all(len(x) == 4 for x in g.ep_model_groups)
    assert all(len(x) == 2 for x in g.ep_data_groups)
    seen = sorted(r for grp in g.ep_model_groups for r in grp)
    assert seen == list(range(16))
    # EP groups stride by tp
    assert g.ep_model_groups[0] == [0, 2, 4, 6]
    assert g.ep_data_groups[0] == [0, 8]


def test_invalid_sizes():
    with pytest.raises(ValueError):
        _mesh(8, tp=3)
    with pytest.raises(ValueError):
        _mesh(8, tp=2, ep=8)


def test_kv_shared_mesh():
    mesh = ps._build_kv_shared_mesh([[0, 1, 2, 3, 4, 5, 6, 7]], 4)
    assert mesh == [[0, 1, 2, 3], [4, 5, 6, 7]]


def test_token_shuffle_mesh():
    mesh = ps._build_token_shuffle_mesh([[0, 2, 4, 6], [1, 3, 5, 7]], 2)
    assert mesh == [[0, 2], [4, 6], [1, 3], [5, 7]]


def test_spec_draft_mesh():
    # draft tp 2 inside tp 4: sub-groups split each tp row
    import neuronx_distributed_amd.parallel.parallel_state as ps
    g = _mesh(8, tp=4)
    # mesh-only mode: build the draft mesh by the same rule used in
    # initialize_speculative_draft_group
    tp_mesh = g.tp_groups
    mesh = []
    for row in tp_mesh:
        for i in range(0, len(row), 2):
            mesh.append(row[i:i + 2])
    assert mesh == [[0, 1], [2, 3], [4, 5], [6, 7]]
