"""UnifiedTransformer / UNIMO prefix-LM families.

Reference behavior: paddlenlp/transformers/{unified_transformer,unimo}/.
"""
import pytest
import torch

from paddlenlp_amd.transformers import (
    UNIMOConfig,
    UNIMOLMHeadModel,
    UnifiedTransformerConfig,
    UnifiedTransformerLMHeadModel,
)
from paddlenlp_amd.transformers.unified_transformer.modeling import (
    prefix_lm_mask,
)

torch.manual_seed(0)

TINY = dict(vocab_size=100, hidden_size=32, num_hidden_layers=2,
            num_attention_heads=4, intermediate_size=64,
            max_position_embeddings=64, hidden_dropout_prob=0.0,
            attention_probs_dropout_prob=0.0)


def test_prefix_mask_shape_and_semantics():
    mask = prefix_lm_mask(4, torch.tensor([2]), past_len=0,
                          device=torch.device("cpu"))
    assert mask.shape == (1, 1, 4, 4)
    m = mask[0, 0]
    # row 0 (query 0): sees itself + prefix cols (<2)
    assert m[0, 0] == 0 and m[0, 1] == 0 and m[0, 2] == float("-inf")
    # row 3: causal, sees everything
    assert (m[3] == 0).all()
    # with past: single query at absolute position past_len
    mask = prefix_lm_mask(1, torch.tensor([2]), past_len=5,
                          device=torch.device("cpu"))
    assert mask.shape == (1, 1, 1, 6) and (mask[0, 0, 0] == 0).all()


@pytest.mark.parametrize("cfg_cls,lm_cls", [
    (UnifiedTransformerConfig, UnifiedTransformerLMHeadModel),
    (UNIMOConfig, UNIMOLMHeadModel),
])
def test_prefix_lm_contract(cfg_cls, lm_cls):
    m = lm_cls(cfg_cls(**TINY)).eval()
    ids = torch.randint(3, 100, (2, 10))
    pl = torch.tensor([6, 4])
    loss, logits = m(ids, prefix_len=pl, labels=ids)
    assert logits.shape == (2, 10, 100)
    loss.backward()
    # cached decode parity (prefix chunk + token-by-token)
    with torch.no_grad():
        full = m(ids, prefix_len=pl)
        lg, past = m(ids[:, :6], prefix_len=pl, use_cache=True)
        outs = [lg]
        for t in range(6, 10):
            lg, past = m(ids[:, t:t + 1], prefix_len=pl,
                         past_key_values=past, use_cache=True)
            outs.append(lg)
    torch.testing.assert_close(torch.cat(outs, 1), full,
                               rtol=2e-4, atol=2e-4)
    out, _ = m.generate(ids[:, :5], max_new_tokens=4)
    assert out.shape[0] == 2 and out.shape[1] <= 4


def test_prefix_is_bidirectional_response_is_causal():
    m = UnifiedTransformerLMHeadModel(
        UnifiedTransformerConfig(**TINY)).eval()
    ids = torch.randint(3, 100, (1, 10))
    ids2 = ids.clone()
    ids2[:, 9] = (ids[:, 9] + 1) % 97 + 3
    with torch.no_grad():
        # whole sequence in the prefix: early positions see the change
        a = m(ids, prefix_len=torch.tensor([10]))
        b = m(ids2, prefix_len=torch.tensor([10]))
        assert not torch.allclose(a[:, 0], b[:, 0])
        # pure causal: they must not
        a = m(ids, prefix_len=torch.tensor([0]))
        b = m(ids2, prefix_len=torch.tensor([0]))
    torch.testing.assert_close(a[:, :9], b[:, :9], rtol=1e-5, atol=1e-5)


def test_unimo_role_embeddings():
    cfg = UNIMOConfig(role_type_size=3, **TINY)
    m = UNIMOLMHeadModel(cfg).eval()
    assert m.unimo is m.unified_transformer
    ids = torch.randint(3, 100, (1, 8))
    roles0 = torch.zeros(1, 8, dtype=torch.long)
    roles1 = torch.ones(1, 8, dtype=torch.long)
    with torch.no_grad():
        a = m(ids, role_ids=roles0)
        b = m(ids, role_ids=roles1)
    assert not torch.allclose(a, b)


def test_chatglm_v1_2d_positions_and_alpha_residual():
    from paddlenlp_amd.transformers import ChatGLMConfig, ChatGLMForCausalLM
    from paddlenlp_amd.transformers.chatglm.modeling import glm_2d_positions

    pos, blk = glm_2d_positions(torch.tensor([4]), 0, 7, torch.device("cpu"))
    assert pos[0].tolist() == [0, 1, 2, 3, 3, 3, 3]
    assert blk[0].tolist() == [0, 0, 0, 0, 1, 2, 3]
    # decode step at absolute index 6 with prefix 4 -> (3, 3)
    pos, blk = glm_2d_positions(torch.tensor([4]), 6, 1, torch.device("cpu"))
    assert pos[0].tolist() == [3] and blk[0].tolist() == [3]

    cfg = ChatGLMConfig(vocab_size=100, hidden_size=32, num_hidden_layers=2,
                        num_attention_heads=4, inner_hidden_size=64,
                        max_sequence_length=64)
    m = ChatGLMForCausalLM(cfg).eval()
    assert abs(m.chatglm.layers[0].alpha - 2.0) < 1e-9  # sqrt(2*2)
    ids = torch.randint(4, 100, (2, 10))
    pl = torch.tensor([6, 5])
    loss, logits = m(ids, prefix_len=pl, labels=ids)
    loss.backward()
    # cached decode parity (chunk covers the longest prefix)
    with torch.no_grad():
        full = m(ids, prefix_len=pl)
        lg, past = m(ids[:, :6], prefix_len=pl, use_cache=True)
        outs = [lg]
        for t in range(6, 10):
            lg, past = m(ids[:, t:t + 1], prefix_len=pl,
                         past_key_values=past, use_cache=True)
            outs.append(lg)
    torch.testing.assert_close(torch.cat(outs, 1), full, rtol=2e-4, atol=2e-4)
    out, _ = m.generate(ids[:, :5], max_new_tokens=4)
    assert out.shape[0] == 2
