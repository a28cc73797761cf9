import torch

from lpp_amd.config import model_config
from lpp_amd.models import LlamaForCausalLM, init_monolithic_weights, loss_fn


def test_monolithic_forward_shapes():
    cfg = model_config("llama-tiny")
    model = LlamaForCausalLM(cfg)
    init_monolithic_weights(model, seed=7)
    ids = torch.randint(0, cfg.vocab_size, (2, 16))
    logits = model(ids)
    assert logits.shape == (2, 16, cfg.vocab_size)
    assert torch.isfinite(logits).all()


def test_loss_and_grads_flow():
    cfg = model_config("llama-tiny")
    model = LlamaForCausalLM(cfg)
    init_monolithic_weights(model, seed=7)
    ids = torch.randint(0, cfg.vocab_size, (2, 16))
    loss = model.compute_loss(ids, ids.clone())
    assert torch.isfinite(loss)
    loss.backward()
    for n, p in model.named_parameters():
        assert p.grad is not None, n
        assert torch.isfinite(p.grad).all(), n


def test_deterministic_init_is_stage_independent():
    cfg = model_config("llama-tiny")
    m1 = LlamaForCausalLM(cfg)
    m2 = LlamaForCausalLM(cfg)
    init_monolithic_weights(m1, seed=3)
    init_monolithic_weights(m2, seed=3)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_activation_checkpointing_same_output():
    cfg = model_config("llama-tiny")
    m1 = LlamaForCausalLM(cfg, activation_checkpointing=False)
    m2 = LlamaForCausalLM(cfg, activation_checkpointing=True)
    init_monolithic_weights(m1, seed=5)
    init_monolithic_weights(m2, seed=5)
    m1.train(); m2.train()
    ids = torch.randint(0, cfg.vocab_size, (2, 16))
    l1 = m1.compute_loss(ids, ids.clone())
    l2 = m2.compute_loss(ids, ids.clone())
    assert torch.allclose(l1, l2, atol=1e-6)
    l1.backward(); l2.backward()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-6)


def test_gqa_presets():
    """LLaMA-2/3 GQA presets build and the kv projections are narrower."""
    from lpp_amd.config import model_config
    from lpp_amd.models import LlamaForCausalLM, loss_fn
    import torch

    mcfg = model_config("llama2-70b", num_layers=1, hidden_size=128, num_heads=8,
                        num_kv_heads=2, intermediate_size=256, vocab_size=128,
                        max_seq_len=64)
    model = LlamaForCausalLM(mcfg)
    attn = model.layers[1].self_attn
    assert attn.k_proj.weight.shape[0] == 2 * (128 // 8)
    ids = torch.randint(0, 128, (2, 32))
    loss = model.compute_loss(ids, ids.clone())
    assert torch.isfinite(loss)


def test_selective_checkpointing_equivalence():
    """Per-layer selective checkpointing must not change the math."""
    import torch
    from lpp_amd.config import model_config
    from lpp_amd.models import LlamaForCausalLM, get_layers_from_config

    mcfg = model_config("llama-tiny", num_layers=4, max_seq_len=32)
    torch.manual_seed(3)
    ref = LlamaForCausalLM(mcfg)
    specs = get_layers_from_config(mcfg, checkpoint_fn=lambda i: i % 2 == 0)
    torch.manual_seed(3)
    sel = torch.nn.ModuleList([sp.build() for sp in specs])
    for a, b in zip(ref.layers.state_dict().values(), sel.state_dict().values()):
        assert torch.equal(a, b)
    ids = torch.randint(0, mcfg.vocab_size, (2, 16))
    ref.train(); x1 = ids
    for l in ref.layers: x1 = l(x1)
    x2 = ids
    for l in sel: l.train(); x2 = l(x2)
    assert torch.allclose(x1, x2, atol=1e-6)
    x2.sum().backward()
    x1.sum().backward()
    g1 = next(ref.layers[1].parameters()).grad
    g2 = next(sel[1].parameters()).grad
    assert torch.allclose(g1, g2, atol=1e-5)


def test_family_a_layers_from_model():
    """Reference family A (get_model :119-125): pipeline layers taken from a
    loaded monolithic model share storage and reproduce its forward."""
    import torch

    from lpp_amd.config import model_config
    from lpp_amd.models import LlamaForCausalLM, layers_from_model, loss_fn
    from lpp_amd.pipeline_module import PipelineModule
    from lpp_amd.topology import ProcessGrid

    mcfg = model_config("llama-tiny", num_layers=2, max_seq_len=32)
    torch.manual_seed(8)
    mono = LlamaForCausalLM(mcfg)
    grid = ProcessGrid(1, 0, 1)
    module = PipelineModule(layers_from_model(mono), grid, loss_fn=loss_fn)
    ids = torch.randint(0, mcfg.vocab_size, (1, 16))
    with torch.no_grad():
        assert torch.equal(module(ids), mono(ids))
    # shared storage, not copies
    assert next(module.layers[1].parameters()).data_ptr() == \
        next(mono.layers[1].parameters()).data_ptr()
