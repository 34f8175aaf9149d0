"""Long-tail decoder families: Gemma, OPT, BLOOM.

Reference behavior: paddlenlp/transformers/{gemma,opt,bloom}/modeling.py.
"""
import torch

from paddlenlp_amd.transformers import (
    BloomConfig,
    BloomForCausalLM,
    GemmaConfig,
    GemmaForCausalLM,
    OPTConfig,
    OPTForCausalLM,
)

torch.manual_seed(0)


def _shifted(ids):
    labels = ids.clone()
    labels[:, :-1] = ids[:, 1:]
    labels[:, -1] = -100
    return labels


def _check_family(model, vocab=128):
    ids = torch.randint(3, vocab, (2, 12))
    loss, logits = model(input_ids=ids, labels=_shifted(ids))
    assert 2.0 < float(loss) < 12.0  # next-token CE at random init
    loss.backward()
    model.zero_grad(set_to_none=True)

    model.eval()
    with torch.no_grad():
        full = model(input_ids=ids)
        _, past = model(input_ids=ids[:, :-1], use_cache=True)
        step, _ = model(input_ids=ids[:, -1:], use_cache=True,
                        past_key_values=past)
    torch.testing.assert_close(step[:, 0], full[:, -1], rtol=1e-4, atol=1e-4)

    out, _ = model.generate(ids[:, :4], max_new_tokens=4, do_sample=False)
    assert out.shape == (2, 4)


def test_gemma():
    cfg = GemmaConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, head_dim=16,
                      max_position_embeddings=64)
    m = GemmaForCausalLM(cfg)
    # gemma quirks: (1+w) RMSNorm with zero-init weight, scaled embeddings,
    # explicit head_dim decoupled from hidden/heads
    assert (m.gemma.layers[0].input_layernorm.weight == 0).all()
    assert m.gemma.embed_scale == 8.0  # sqrt(64)
    assert cfg.head_dim == 16 and cfg.hidden_size // cfg.num_attention_heads == 16 \
        or cfg.head_dim == 16
    assert m.lm_head.weight.data_ptr() == m.gemma.embed_tokens.weight.data_ptr()
    _check_family(m)


def test_gemma_head_dim_decoupled():
    cfg = GemmaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=2,
                      num_key_value_heads=1, head_dim=24,  # != 32/2
                      max_position_embeddings=32)
    m = GemmaForCausalLM(cfg)
    attn = m.gemma.layers[0].self_attn
    assert attn.head_dim == 24
    ids = torch.randint(3, 64, (1, 8))
    logits = m(input_ids=ids)
    assert logits.shape == (1, 8, 64)


def test_gemma_config_roundtrip(tmp_path):
    cfg = GemmaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=2,
                      num_key_value_heads=1, head_dim=24,
                      max_position_embeddings=32)
    cfg.save_pretrained(str(tmp_path))
    cfg2 = GemmaConfig.from_pretrained(str(tmp_path))
    assert cfg2.head_dim == 24


def test_opt():
    cfg = OPTConfig(vocab_size=128, hidden_size=64, num_hidden_layers=2,
                    num_attention_heads=4, intermediate_size=128,
                    max_position_embeddings=64)
    m = OPTForCausalLM(cfg)
    # +2 reserved position rows
    assert m.opt.embed_positions.num_embeddings == 66
    _check_family(m)


def test_bloom():
    cfg = BloomConfig(vocab_size=128, hidden_size=64, num_hidden_layers=2,
                      num_attention_heads=4)
    m = BloomForCausalLM(cfg)
    # no positional embeddings anywhere: alibi carries position
    assert not any("position" in n for n, _ in m.named_parameters())
    _check_family(m)


def test_bloom_alibi_monotone():
    """Farther keys get a more negative bias (per head slope)."""
    m = BloomForCausalLM(BloomConfig(vocab_size=64, hidden_size=32,
                                     num_hidden_layers=1, num_attention_heads=4))
    bias = m.bloom._alibi(8, torch.device("cpu"))
    assert bias.shape == (1, 4, 1, 8)
    assert (bias[0, :, 0, 1:] <= bias[0, :, 0, :-1]).all()


def test_falcon_parallel_residual():
    from paddlenlp_amd.transformers import FalconConfig, FalconForCausalLM

    cfg = FalconConfig(vocab_size=128, hidden_size=64, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=1,
                       intermediate_size=128, max_position_embeddings=64)
    m = FalconForCausalLM(cfg)
    # parallel form: single LN per layer (no post_attention_layernorm)
    assert not hasattr(m.falcon.h[0], "post_attention_layernorm")
    _check_family(m)

    # sequential form has both norms
    cfg2 = FalconConfig(vocab_size=128, hidden_size=64, num_hidden_layers=1,
                        num_attention_heads=4, num_key_value_heads=2,
                        intermediate_size=128, max_position_embeddings=64,
                        parallel_attn=False)
    m2 = FalconForCausalLM(cfg2)
    assert hasattr(m2.falcon.h[0], "post_attention_layernorm")
    _check_family(m2)


def test_chatglm_v2_partial_rope():
    from paddlenlp_amd.transformers import ChatGLMv2Config, ChatGLMv2ForCausalLM
    from paddlenlp_amd.transformers.chatglm_v2.modeling import _glm_rope

    cfg = ChatGLMv2Config(vocab_size=128, hidden_size=64, num_hidden_layers=2,
                          num_attention_heads=4, multi_query_group_num=2,
                          ffn_hidden_size=128, kv_channels=16,
                          max_position_embeddings=64)
    m = ChatGLMv2ForCausalLM(cfg)
    # rotary touches only the first half of the head dim
    x = torch.randn(1, 4, 2, 16)
    n = 16 // 4
    inv = 1.0 / (10000.0 ** (torch.arange(n).float() / n))
    freqs = torch.outer(torch.arange(4).float(), inv)
    out = _glm_rope(x, freqs.cos(), freqs.sin())
    torch.testing.assert_close(out[..., 8:], x[..., 8:])       # pass-through
    assert not torch.allclose(out[..., :8], x[..., :8])        # rotated
    # position 0 is identity everywhere
    torch.testing.assert_close(out[:, 0], x[:, 0])
    _check_family(m)


def test_gptj_partial_rope_parallel_residual():
    from paddlenlp_amd.transformers import GPTJConfig, GPTJForCausalLM
    from paddlenlp_amd.transformers.gptj.modeling import _gptj_rope

    cfg = GPTJConfig(vocab_size=128, hidden_size=64, num_hidden_layers=2,
                     num_attention_heads=4, rotary_dim=8,
                     intermediate_size=128, max_position_embeddings=64)
    m = GPTJForCausalLM(cfg)
    # untied head with bias
    assert m.lm_head.bias is not None
    assert m.lm_head.weight.data_ptr() != m.gptj.wte.weight.data_ptr()
    # rope leaves dims >= rotary_dim untouched; position 0 is identity
    x = torch.randn(1, 4, 2, 16)
    n = 8 // 2
    inv = 1.0 / (10000.0 ** (torch.arange(n).float() * 2 / 8))
    freqs = torch.outer(torch.arange(4).float(), inv)
    out = _gptj_rope(x, freqs.cos(), freqs.sin(), 8)
    torch.testing.assert_close(out[..., 8:], x[..., 8:])
    torch.testing.assert_close(out[:, 0], x[:, 0])
    _check_family(m)
