from lpp_amd.config import model_config
from lpp_amd.layer_spec import LayerSpec, partition_balanced, partition_uniform
from lpp_amd.models import get_layers_from_config


def _max_part(weights, bounds):
    return max(sum(weights[bounds[s] : bounds[s + 1]]) for s in range(len(bounds) - 1))


def test_balanced_simple():
    w = [1, 1, 1, 1]
    assert partition_balanced(w, 2) == [0, 2, 4]


def test_balanced_skewed():
    # big embedding + head dominate
    w = [100, 10, 10, 10, 10, 100]
    b = partition_balanced(w, 3)
    assert b[0] == 0 and b[-1] == 6
    # brute force optimality check
    best = min(
        max(sum(w[:i]), sum(w[i:j]), sum(w[j:]))
        for i in range(1, 5)
        for j in range(i + 1, 6)
    )
    assert _max_part(w, b) == best


def test_every_stage_nonempty():
    w = [5] * 11
    b = partition_balanced(w, 4)
    for s in range(4):
        assert b[s + 1] > b[s]


def test_uniform():
    assert partition_uniform(10, 4) == [0, 3, 6, 8, 10]


def test_llama_spec_weights():
    cfg = model_config("llama-65b")
    specs = get_layers_from_config(cfg)
    assert len(specs) == cfg.num_layers + 3
    weights = [s.param_count() for s in specs]
    # embedding and head = vocab*hidden
    assert weights[0] == cfg.vocab_size * cfg.hidden_size
    assert weights[-1] == cfg.vocab_size * cfg.hidden_size
    assert weights[-2] == cfg.hidden_size
    # total matches config param count
    assert sum(weights) == cfg.num_params()
    bounds = partition_balanced(weights, 8)
    parts = [sum(weights[bounds[s] : bounds[s + 1]]) for s in range(8)]
    assert max(parts) / min(parts) < 1.35  # reasonably balanced for 83 layers / 8 stages


def test_65b_pp8_partition_balanced():
    """The headline N=8 partition: embed rides with stage 0, norm+head with
    stage 7, decoder layers 10-11 per stage, param spread < 4%."""
    from lpp_amd.config import model_config
    from lpp_amd.layer_spec import partition_balanced
    from lpp_amd.models import get_layers_from_config

    m = model_config("llama-65b", num_layers=80, max_seq_len=4096)
    specs = get_layers_from_config(m)
    w = [s.param_count() for s in specs]
    bounds = partition_balanced(w, 8)
    per_stage = [sum(w[bounds[i]:bounds[i + 1]]) for i in range(8)]
    assert bounds[0] == 0 and bounds[-1] == len(specs) == 83
    assert max(per_stage) / min(per_stage) < 1.04
    n_specs = [bounds[i + 1] - bounds[i] for i in range(8)]
    assert all(10 <= n <= 12 for n in n_specs)
