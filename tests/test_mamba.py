"""Mamba selective-SSM family: scan correctness, decode-cache parity.

Reference behavior: paddlenlp/transformers/mamba/modeling.py.
"""
import torch

from paddlenlp_amd.transformers import MambaConfig, MambaForCausalLM
from paddlenlp_amd.transformers.mamba.modeling import MambaMixer

torch.manual_seed(0)


def tiny_cfg(**kw):
    return MambaConfig(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                       state_size=8, conv_kernel=4, expand=2, **kw)


def test_mixer_shapes_and_state():
    mix = MambaMixer(tiny_cfg())
    x = torch.randn(2, 10, 32)
    out, (conv_state, ssm_state) = mix(x)
    assert out.shape == (2, 10, 32)
    assert conv_state.shape == (2, 64, 3)   # D = expand*h, K-1
    assert ssm_state.shape == (2, 64, 8)
    # A starts negative (stable recurrence)
    assert (-torch.exp(mix.A_log) < 0).all()


def test_incremental_decode_matches_full_scan():
    """Stepping one token at a time through the (conv, ssm) cache must equal
    the full-sequence scan."""
    mix = MambaMixer(tiny_cfg()).eval()
    x = torch.randn(1, 8, 32)
    with torch.no_grad():
        full, _ = mix(x)
        cache = (torch.zeros(1, 64, 3), torch.zeros(1, 64, 8))
        steps = []
        for t in range(8):
            y, cache = mix(x[:, t:t + 1], cache)
            steps.append(y)
    inc = torch.cat(steps, dim=1)
    torch.testing.assert_close(inc, full, rtol=1e-4, atol=1e-5)


def test_mamba_lm_train_and_generate():
    m = MambaForCausalLM(tiny_cfg())
    ids = torch.randint(3, 96, (2, 12))
    labels = ids.clone()
    labels[:, :-1] = ids[:, 1:]
    labels[:, -1] = -100
    loss, logits = m(input_ids=ids, labels=labels)
    assert 2.0 < float(loss) < 12.0
    loss.backward()
    assert m.mamba.layers[0].mixer.A_log.grad is not None
    assert m.mamba.layers[0].mixer.dt_proj.bias.grad is not None

    m.eval()
    with torch.no_grad():
        full = m(input_ids=ids)
        _, cache = m(input_ids=ids[:, :-1], use_cache=True)
        step, _ = m(input_ids=ids[:, -1:], use_cache=True, past_key_values=cache)
    torch.testing.assert_close(step[:, 0], full[:, -1], rtol=1e-4, atol=1e-4)

    out, _ = m.generate(ids[:, :4], max_new_tokens=4, do_sample=False)
    assert out.shape == (2, 4)


def test_mamba_save_load(tmp_path):
    m = MambaForCausalLM(tiny_cfg()).eval()
    m.save_pretrained(str(tmp_path))
    m2 = MambaForCausalLM.from_pretrained(str(tmp_path)).eval()
    ids = torch.randint(3, 96, (1, 6))
    with torch.no_grad():
        torch.testing.assert_close(m(input_ids=ids), m2(input_ids=ids))


def test_jamba_hybrid_layout_and_parity():
    from paddlenlp_amd.transformers import JambaConfig, JambaForCausalLM
    from paddlenlp_amd.transformers.jamba.modeling import (
        JambaAttention,
        JambaMoE,
    )
    from paddlenlp_amd.transformers.mamba.modeling import MambaMixer

    torch.manual_seed(0)
    cfg = JambaConfig(vocab_size=96, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=4, num_attention_heads=4,
                      num_key_value_heads=2, attn_layer_period=4,
                      attn_layer_offset=1, expert_layer_period=2,
                      expert_layer_offset=0, num_experts=4,
                      num_experts_per_tok=2, mamba_d_state=8,
                      max_position_embeddings=64)
    m = JambaForCausalLM(cfg)
    layers = m.jamba.layers
    # layer 1 is attention, the rest mamba; layers 0 and 2 are MoE
    assert isinstance(layers[1].mixer, JambaAttention)
    assert isinstance(layers[0].mixer, MambaMixer)
    assert isinstance(layers[0].feed_forward, JambaMoE)
    assert not layers[1].is_moe

    ids = torch.randint(3, 96, (2, 12))
    labels = ids.clone()
    labels[:, :-1] = ids[:, 1:]
    labels[:, -1] = -100
    loss, logits = m(input_ids=ids, labels=labels)
    assert 2.0 < float(loss) < 12.0
    loss.backward()
    assert layers[0].feed_forward.router.weight.grad is not None

    # cached decode parity across the mixed cache types
    m.eval()
    with torch.no_grad():
        full = m(input_ids=ids)
        _, past = m(input_ids=ids[:, :-1], use_cache=True)
        step, _ = m(input_ids=ids[:, -1:], use_cache=True, past_key_values=past)
    torch.testing.assert_close(step[:, 0], full[:, -1], rtol=1e-4, atol=1e-4)

    out, _ = m.generate(ids[:, :4], max_new_tokens=4, do_sample=False)
    assert out.shape == (2, 4)
