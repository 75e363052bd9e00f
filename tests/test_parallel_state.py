"""Rank-grid arithmetic vs reference semantics (no GPU needed)."""

from megatronapp_amd.core.parallel_state import RankGenerator


def test_tp_fastest_varying():
    gen = RankGenerator(tp=2, dp=2, pp=2, cp=1)
    tp_groups = gen.get_ranks("tp")
    assert [0, 1] in tp_groups and [2, 3] in tp_groups
    assert len(tp_groups) == 4


def test_pp_slowest_varying():
    gen = RankGenerator(tp=2, dp=2, pp=2, cp=1)
    pp_groups = gen.get_ranks("pp")
    assert [0, 4] in pp_groups
    assert [3, 7] in pp_groups


def test_dp_groups_tp2_pp2():
    gen = RankGenerator(tp=2, dp=2, pp=2, cp=1)
    dp_groups = gen.get_ranks("dp")
    # dp strides over tp*cp
    assert [0, 2] in dp_groups and [1, 3] in dp_groups
    assert [4, 6] in dp_groups and [5, 7] in dp_groups


def test_cp_between_tp_and_dp():
    gen = RankGenerator(tp=2, dp=2, pp=1, cp=2)
    cp_groups = gen.get_ranks("cp")
    assert [0, 2] in cp_groups and [1, 3] in cp_groups
    dp_cp = gen.get_ranks("dp-cp")
    assert [0, 2, 4, 6] in dp_cp


def test_model_parallel_group():
    gen = RankGenerator(tp=2, dp=2, pp=2, cp=1)
    mp_groups = gen.get_ranks("tp-pp")
    assert [0, 1, 4, 5] in mp_groups
    assert [2, 3, 6, 7] in mp_groups


def test_world_size_product():
    gen = RankGenerator(tp=4, dp=2, pp=3, cp=2)
    assert gen.world_size == 48
    for token in ("tp", "dp", "pp", "cp"):
        groups = gen.get_ranks(token)
        seen = sorted(r for g in groups for r in g)
        assert seen == list(range(48))
