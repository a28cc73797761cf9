"""Multi-process pipeline tests on gloo (BASELINE config #1: plumbing on CPU,
world_size=2 — SURVEY.md §4 'Schedule/deadlock tests without GPUs')."""

import pytest
import torch

from tests.dist_utils import run_dist
from tests.engine_utils import run_steps


def _single_process_baseline(steps=3, gas=4):
    return run_steps(rank=0, world_size=1, num_stages=1, steps=steps, gas=gas)


def test_pp2_matches_single_process():
    """PP=2 over gloo must reproduce the monolithic loss trajectory exactly
    (fp32, identical per-layer weights + data order)."""
    base = _single_process_baseline()
    got = run_dist(2, run_steps, 2, 3, 4)  # num_stages=2, steps=3, gas=4
    for r in range(2):
        for a, b in zip(base, got[r]):
            assert abs(a - b) < 1e-4, (base, got[r])


def test_pp2_more_microbatches_than_stages():
    """gas > stages exercises the steady-state 1F1B path."""
    base = _single_process_baseline(steps=2, gas=6)
    got = run_dist(2, run_steps, 2, 2, 6)
    for a, b in zip(base, got[0]):
        assert abs(a - b) < 1e-4


def test_pp2_fewer_microbatches_than_depth():
    """gas=1: pure fill-drain, no steady state (warmup==M edge case)."""
    base = _single_process_baseline(steps=2, gas=1)
    got = run_dist(2, run_steps, 2, 2, 1)
    for a, b in zip(base, got[0]):
        assert abs(a - b) < 1e-4


def test_dp2_pure_data_parallel():
    """PP=1 x DP=2: both ranks return the identical (dp-averaged) loss and
    stay in sync across steps."""
    got = run_dist(2, run_steps, 1, 3, 4)
    assert got[0] == pytest.approx(got[1], abs=1e-6)


def test_pp2_all_ranks_same_loss():
    got = run_dist(2, run_steps, 2, 2, 4)
    assert got[0] == pytest.approx(got[1], abs=1e-6)


def test_hybrid_pp2_dp2_matches_single_process():
    """world 4 = PP2 x DP2 (the reference's hybrid mode, README.md:39-42):
    same example set per step as a single process with gas doubled, so the
    loss trajectory must match within fp tolerance."""
    base = _single_process_baseline(steps=3, gas=8)
    got = run_dist(4, run_steps, 2, 3, 4)  # stages=2 -> dp=2, gas=4/rank
    for r in range(4):
        for a, b in zip(base, got[r]):
            assert abs(a - b) < 1e-3, (base, got[r])


def test_hybrid_all_ranks_agree():
    got = run_dist(4, run_steps, 2, 2, 4)
    for r in range(1, 4):
        assert got[0] == pytest.approx(got[r], abs=1e-6)


def _eval_run(rank, world, steps=1):
    from tests.engine_utils import make_config, build_engine, sequential_loader

    cfg = make_config(num_stages=world, gas=4)
    engine = build_engine(cfg, rank, world)
    it = sequential_loader(cfg) if (engine.is_first_stage or engine.is_last_stage) else None
    return float(engine.eval_batch(it, micro_batches=4))


def test_eval_batch_pp2():
    """Forward-only eval over a 2-stage pipeline: both ranks get the loss."""
    got = run_dist(2, _eval_run)
    assert got[0] == pytest.approx(got[1], abs=1e-6)
    assert got[0] > 0


def test_pp4_matches_single_process():
    """4-stage pipeline (deep warmup/cooldown, the N=8 schedule shape in
    miniature) reproduces the monolithic trajectory."""
    base = _single_process_baseline(steps=2, gas=8)
    got = run_dist(4, run_steps, 4, 2, 8)  # stages=4
    for r in range(4):
        for a, b in zip(base, got[r]):
            assert abs(a - b) < 1e-3, (base, got[r])


def test_pp4_gas_less_than_stages():
    """gas=2 < stages=4: some stages do zero steady-state iterations."""
    base = _single_process_baseline(steps=2, gas=2)
    got = run_dist(4, run_steps, 4, 2, 2)
    for a, b in zip(base, got[0]):
        assert abs(a - b) < 1e-3


def test_p2p_overlap_off_matches_on():
    """The serial (p2p_overlap=False) and overlapped schedules are the same
    numerics — bit-identical loss trajectories."""
    on = run_dist(2, run_steps, 2, 2, 6)
    off = run_dist(2, run_steps, 2, 2, 6, "fp32", 11, False)
    assert on[0] == off[0]


def test_overlap_allreduce_off_matches_on():
    """Bucket all-reduces launched from the final backward's hooks produce
    the same trajectory as the boundary all-reduce (PP2 x DP2)."""
    on = run_dist(4, run_steps, 2, 2, 4)
    off = run_dist(4, run_steps, 2, 2, 4, "fp32", 11, True, False)
    assert on[0] == pytest.approx(off[0], abs=1e-6)


@pytest.mark.slow
def test_pp8_matches_single_process():
    """world 8 = the driver's N=8 scale-run schedule shape, on gloo."""
    base = _single_process_baseline(steps=2, gas=8)
    got = run_dist(8, run_steps, 8, 2, 8, timeout=600.0)
    for a, b in zip(base, got[0]):
        assert abs(a - b) < 1e-3, (base, got[0])


def test_pp3_matches_single_process():
    """Odd stage count: stage 1 is a pure middle stage under the overlapped
    schedule (pre-posted recvs on both channels simultaneously)."""
    base = _single_process_baseline(steps=2, gas=6)
    got = run_dist(3, run_steps, 3, 2, 6)
    for r in range(3):
        for a, b in zip(base, got[r]):
            assert abs(a - b) < 1e-3, (base, got[r])


@pytest.mark.slow
@pytest.mark.parametrize("world,gas", [(3, 2), (3, 4), (5, 3), (5, 7), (6, 6)])
def test_schedule_property_sweep(world, gas):
    """Randomized-shape sweep of the overlapped 1F1B schedule: every
    (depth, microbatch-count) combination — including gas < depth and
    non-power-of-two depths — must reproduce the monolithic trajectory."""
    base = _single_process_baseline(steps=2, gas=gas)
    got = run_dist(world, run_steps, world, 2, gas, timeout=420.0)
    for a, b in zip(base, got[0]):
        assert abs(a - b) < 1e-3, (world, gas, base, got[0])
