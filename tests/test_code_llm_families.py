"""QWen v1 + CodeGen family specifics (matrix covers the shared contract).

Reference behavior: paddlenlp/transformers/{qwen,codegen}/modeling.py.
"""
import torch

from paddlenlp_amd.transformers import (
    CodeGenConfig,
    CodeGenForCausalLM,
    QWenConfig,
    QWenForCausalLM,
    QWenLMHeadModel,
)

torch.manual_seed(0)


def test_qwen_structure():
    cfg = QWenConfig(vocab_size=96, hidden_size=32, intermediate_size=128,
                     num_hidden_layers=2, num_attention_heads=4,
                     seq_length=32, max_position_embeddings=64)
    m = QWenForCausalLM(cfg)
    blk = m.qwen.h[0]
    # c_attn always biased; everything else follows no_bias=True
    assert blk.attn.c_attn.bias is not None
    assert blk.attn.c_proj.bias is None
    assert blk.mlp.w1.bias is None
    # split-half swiglu: ff = intermediate // 2
    assert blk.mlp.w1.out_features == 64
    assert QWenLMHeadModel is QWenForCausalLM


def test_qwen_mlp_matches_reference_formula():
    cfg = QWenConfig(vocab_size=96, hidden_size=32, intermediate_size=128,
                     num_hidden_layers=1, num_attention_heads=4,
                     no_bias=False, seq_length=32, max_position_embeddings=64)
    mlp = QWenForCausalLM(cfg).qwen.h[0].mlp
    x = torch.randn(2, 5, 32)
    with torch.no_grad():
        want = mlp.c_proj(mlp.w1(x) * torch.nn.functional.silu(mlp.w2(x)))
        torch.testing.assert_close(mlp(x), want, rtol=1e-5, atol=1e-5)


def test_qwen_logn_scaling_beyond_train_length():
    cfg = QWenConfig(vocab_size=96, hidden_size=32, intermediate_size=128,
                     num_hidden_layers=2, num_attention_heads=4,
                     seq_length=8, use_logn_attn=True,
                     max_position_embeddings=64)
    m = QWenForCausalLM(cfg).eval()
    ids = torch.randint(3, 96, (1, 16))
    with torch.no_grad():
        a = m(ids)
        m.qwen.h[0].attn.use_logn_attn = False
        m.qwen.h[1].attn.use_logn_attn = False
        b = m(ids)
    # within the train length logits agree; beyond it logn changes them
    torch.testing.assert_close(a[:, :7], b[:, :7], rtol=1e-4, atol=1e-4)
    assert not torch.allclose(a[:, -1], b[:, -1])


def test_codegen_attribute_map_and_head_bias():
    cfg = CodeGenConfig(vocab_size=96, n_embd=32, n_layer=2, n_head=4,
                        rotary_dim=4, max_position_embeddings=64)
    assert cfg.hidden_size == 32 and cfg.num_hidden_layers == 2
    assert cfg.n_inner == 128  # default 4x
    m = CodeGenForCausalLM(cfg)
    assert m.lm_head.bias is not None
    assert m.transformer.h[0].attn.qkv_proj.bias is None


def test_codegen_partial_rotary():
    # only the first rotary_dim dims rotate: zeroing the passthrough dims of
    # k at one position must not change attention FROM other positions' rope
    from paddlenlp_amd.transformers.gptj.modeling import _gptj_rope
    x = torch.randn(1, 3, 2, 8)
    cos = torch.randn(3, 2)
    sin = torch.randn(3, 2)
    out = _gptj_rope(x, cos, sin, 4)
    torch.testing.assert_close(out[..., 4:], x[..., 4:])
    assert not torch.allclose(out[..., :4], x[..., :4])


def test_yuan_lfa_memory_exactness():
    """Token-by-token decode must match full forward — this exercises the
    LFA 2-token raw-hidden memory carried in the cache's third slot."""
    from paddlenlp_amd.transformers import YuanConfig, YuanForCausalLM

    m = YuanForCausalLM(YuanConfig(
        vocab_size=96, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64)).eval()
    ids = torch.randint(3, 96, (2, 12))
    with torch.no_grad():
        full = m(input_ids=ids)
        past, outs = None, []
        for t in range(ids.shape[1]):
            lg, past = m(input_ids=ids[:, t:t + 1], use_cache=True,
                         past_key_values=past)
            outs.append(lg[:, 0])
    torch.testing.assert_close(torch.stack(outs, 1), full, rtol=2e-4, atol=2e-4)
    # cache carries (k, v, memory[B,2,H])
    assert len(past[0]) == 3 and past[0][2].shape == (2, 2, 32)


def test_yuan_v_reads_raw_hidden():
    """V must come from raw hidden states, Q/K from the filtered path."""
    from paddlenlp_amd.transformers import YuanConfig, YuanForCausalLM

    m = YuanForCausalLM(YuanConfig(
        vocab_size=96, hidden_size=32, intermediate_size=64,
        num_hidden_layers=1, num_attention_heads=4, num_key_value_heads=4,
        max_position_embeddings=64)).eval()
    attn = m.yuan.layers[0].self_attn
    x = torch.randn(1, 5, 32)
    captured = {}
    orig = attn.v_proj.forward

    def spy(inp):
        captured["v_in"] = inp
        return orig(inp)

    attn.v_proj.forward = spy
    with torch.no_grad():
        attn(x)
    torch.testing.assert_close(captured["v_in"], x)


def test_codegen_qkv_conversion_roundtrip():
    """Blocked (mp_num=4, [q_i; v_i; k_i] per block) -> plain [q; k; v]."""
    from paddlenlp_amd.transformers.conversion_utils import (
        _unblock_codegen_qkv,
        convert_hf_state_dict,
    )

    h, mp = 32, 4
    q = torch.randn(h, h)
    k = torch.randn(h, h)
    v = torch.randn(h, h)
    piece = h // mp
    blocks = []
    for i in range(mp):
        blocks += [q[i * piece:(i + 1) * piece],
                   v[i * piece:(i + 1) * piece],
                   k[i * piece:(i + 1) * piece]]
    blocked = torch.cat(blocks, dim=0)
    torch.testing.assert_close(_unblock_codegen_qkv(blocked),
                               torch.cat([q, k, v], dim=0))

    cfg = CodeGenConfig(vocab_size=96, n_embd=h, n_layer=1, n_head=4,
                        rotary_dim=4, max_position_embeddings=64)
    sd = {"transformer.h.0.attn.qkv_proj.weight": blocked,
          "transformer.wte.weight": torch.randn(96, h)}
    out = convert_hf_state_dict(sd, cfg)
    torch.testing.assert_close(out["transformer.h.0.attn.qkv_proj.weight"],
                               torch.cat([q, k, v], dim=0))
    torch.testing.assert_close(out["transformer.wte.weight"],
                               sd["transformer.wte.weight"])
