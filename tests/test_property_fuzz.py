"""Property-based fuzzing (hypothesis) of load-bearing pure logic: the
stage partitioner that decides pp8 layer ownership, the completion-label
masking, and the length-padding batch merge."""

import torch
from hypothesis import given, settings, strategies as st

from lpp_amd.layer_spec import partition_balanced, partition_uniform


def _brute_min_max(weights, stages):
    """Exponential reference: minimal max-part-sum over all contiguous
    splits (only used at tiny sizes)."""
    n = len(weights)
    best = [float("inf")]

    def rec(i, parts_left, cur_max):
        if parts_left == 0:
            if i == n:
                best[0] = min(best[0], cur_max)
            return
        for j in range(i + 1, n - parts_left + 2):
            s = sum(weights[i:j])
            if max(cur_max, s) < best[0]:
                rec(j, parts_left - 1, max(cur_max, s))

    rec(0, stages, 0)
    return best[0]


@settings(max_examples=200, deadline=None)
@given(st.lists(st.integers(min_value=1, max_value=1000), min_size=1, max_size=9),
       st.integers(min_value=1, max_value=9))
def test_partition_balanced_is_optimal(weights, stages):
    if stages > len(weights):
        return
    bounds = partition_balanced(weights, stages)
    # structural invariants
    assert bounds[0] == 0 and bounds[-1] == len(weights)
    assert all(b2 > b1 for b1, b2 in zip(bounds, bounds[1:]))  # >=1 layer/stage
    got = max(sum(weights[bounds[s]:bounds[s + 1]]) for s in range(stages))
    assert got == _brute_min_max(weights, stages)


@settings(max_examples=100, deadline=None)
@given(st.integers(min_value=1, max_value=64), st.integers(min_value=1, max_value=8))
def test_partition_uniform_covers(n, stages):
    if stages > n:
        return
    bounds = partition_uniform(n, stages)
    assert bounds[0] == 0 and bounds[-1] == n
    sizes = [b2 - b1 for b1, b2 in zip(bounds, bounds[1:])]
    assert min(sizes) >= 1 and max(sizes) - min(sizes) <= 1


@settings(max_examples=200, deadline=None)
@given(st.integers(min_value=1, max_value=6),
       st.integers(min_value=1, max_value=20),
       st.integers(min_value=0, max_value=32000),
       st.data())
def test_completion_labels_properties(bsz, seqlen, pad_id, data):
    from lpp_amd.data.text import completion_labels

    ids = torch.randint(0, 32000, (bsz, seqlen))
    lengths = torch.tensor(
        [data.draw(st.integers(min_value=0, max_value=seqlen)) for _ in range(bsz)])
    prompt_lens = torch.tensor(
        [data.draw(st.integers(min_value=0, max_value=int(l))) for l in lengths])
    labels = completion_labels(ids, prompt_lens, pad_id, lengths=lengths)
    for b in range(bsz):
        for t in range(seqlen):
            keep = prompt_lens[b] <= t < lengths[b]
            if keep:
                assert labels[b, t] == ids[b, t], (b, t)
            else:
                assert labels[b, t] == -100, (b, t)


@settings(max_examples=100, deadline=None)
@given(st.integers(min_value=1, max_value=5), st.integers(min_value=1, max_value=9),
       st.integers(min_value=1, max_value=5), st.integers(min_value=1, max_value=9))
def test_combine_on_length_properties(r1, l1, r2, l2):
    from lpp_amd.data import combine_on_length

    a = torch.arange(r1 * l1).reshape(r1, l1)
    b = torch.arange(r2 * l2).reshape(r2, l2) + 1000
    out = combine_on_length(a, b, pad_value=-7)
    L = max(l1, l2)
    assert out.shape == (r1 + r2, L)
    assert torch.equal(out[:r1, :l1], a)
    assert torch.equal(out[r1:, :l2], b)
    if l1 < L:
        assert (out[:r1, l1:] == -7).all()
    if l2 < L:
        assert (out[r1:, l2:] == -7).all()
