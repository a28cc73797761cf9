"""Pipeline-parallel generation == monolithic generation (gloo oracle)."""

import pytest
import torch

from tests.dist_utils import run_dist


def _pp_generate(rank, world, max_new=6):
    from lpp_amd.config import model_config
    from lpp_amd.inference import pipeline_generate
    from lpp_amd.models import get_layers_from_config, init_pipeline_weights, loss_fn
    from lpp_amd.pipeline_module import PipelineModule
    from lpp_amd.topology import ProcessGrid

    mcfg = model_config("llama-tiny", num_layers=4, max_seq_len=64)
    grid = ProcessGrid(world, rank, world)
    grid.build_groups()
    module = PipelineModule(get_layers_from_config(mcfg), grid, loss_fn=loss_fn,
                            device=torch.device("cpu"), dtype=torch.float32)
    init_pipeline_weights(module, mcfg, seed=21)
    g = torch.Generator().manual_seed(77)
    ids = torch.randint(0, mcfg.vocab_size, (2, 5), generator=g)
    out = pipeline_generate(module, grid, ids, max_new_tokens=max_new)
    return out.tolist()


def _mono_generate(max_new=6):
    from lpp_amd.config import model_config
    from lpp_amd.models import LlamaForCausalLM, init_monolithic_weights

    mcfg = model_config("llama-tiny", num_layers=4, max_seq_len=64)
    m = LlamaForCausalLM(mcfg)
    init_monolithic_weights(m, seed=21)
    g = torch.Generator().manual_seed(77)
    ids = torch.randint(0, mcfg.vocab_size, (2, 5), generator=g)
    return m.generate(ids, max_new_tokens=max_new).tolist()


@pytest.mark.parametrize("world", [2, 4])
def test_pipeline_generate_matches_monolithic(world):
    ref = _mono_generate()
    got = run_dist(world, _pp_generate)
    for r in range(world):
        assert got[r] == ref, (got[r], ref)
