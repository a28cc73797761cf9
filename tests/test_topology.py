import pytest

from lpp_amd.topology import ProcessGrid


def test_stage_major_layout():
    # PP=4 x DP=2 on 8 ranks
    g = ProcessGrid(world_size=8, rank=5, num_stages=4)
    assert g.dp_degree == 2
    assert g.stage_id == 2
    assert g.dp_id == 1
    assert g.prev_rank == 3
    assert g.next_rank == 7
    assert not g.is_first_stage() and not g.is_last_stage()


def test_first_last():
    g0 = ProcessGrid(8, 0, 4)
    assert g0.is_first_stage() and g0.prev_rank is None
    gl = ProcessGrid(8, 7, 4)
    assert gl.is_last_stage() and gl.next_rank is None
    assert gl.get_data_parallel_id() == 1


def test_pure_pp():
    for r in range(8):
        g = ProcessGrid(8, r, 8)
        assert g.stage_id == r and g.dp_id == 0


def test_invalid_world():
    with pytest.raises(ValueError):
        ProcessGrid(6, 0, 4)
