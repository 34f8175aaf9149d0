"""One matrix over every registered causal-LM family: tiny build,
next-token loss, backward, cached-decode parity, greedy generate.

Catches registry/forward-contract regressions centrally; family-specific
behavior lives in the per-family test files.
"""
import pytest
import torch

from paddlenlp_amd.transformers.auto.registry import MODEL_REGISTRY, get_class

TINY = dict(
    llama=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
               num_hidden_layers=2, num_attention_heads=4,
               num_key_value_heads=2, max_position_embeddings=64),
    gpt=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
             num_hidden_layers=2, num_attention_heads=4,
             max_position_embeddings=64),
    gptj=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
              num_hidden_layers=2, num_attention_heads=4, rotary_dim=4,
              max_position_embeddings=64),
    qwen2=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
               num_hidden_layers=2, num_attention_heads=4,
               num_key_value_heads=2, max_position_embeddings=64),
    qwen2_moe=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                   num_hidden_layers=2, num_attention_heads=4,
                   num_key_value_heads=2, num_experts=4, num_experts_per_tok=2,
                   moe_intermediate_size=32,
                   shared_expert_intermediate_size=64,
                   max_position_embeddings=64),
    mistral=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                 num_hidden_layers=2, num_attention_heads=4,
                 num_key_value_heads=2, max_position_embeddings=64),
    mixtral=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                 num_hidden_layers=2, num_attention_heads=4,
                 num_key_value_heads=2, num_local_experts=4,
                 num_experts_per_tok=2, max_position_embeddings=64),
    deepseek_v2=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                     moe_intermediate_size=32, num_hidden_layers=2,
                     num_attention_heads=4, q_lora_rank=16, kv_lora_rank=8,
                     qk_nope_head_dim=8, qk_rope_head_dim=4, v_head_dim=8,
                     n_routed_experts=4, n_shared_experts=1,
                     num_experts_per_tok=2, n_group=2, topk_group=1,
                     first_k_dense_replace=1, max_position_embeddings=64),
    gemma=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
               num_hidden_layers=2, num_attention_heads=4,
               num_key_value_heads=2, head_dim=8,
               max_position_embeddings=64),
    opt=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
             num_hidden_layers=2, num_attention_heads=4,
             max_position_embeddings=64),
    bloom=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
               num_attention_heads=4),
    falcon=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                num_hidden_layers=2, num_attention_heads=4,
                num_key_value_heads=1, max_position_embeddings=64),
    chatglm_v2=dict(vocab_size=96, hidden_size=32, ffn_hidden_size=64,
                    num_hidden_layers=2, num_attention_heads=4,
                    multi_query_group_num=2, kv_channels=8,
                    max_position_embeddings=64),
    mamba=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
               state_size=8),
    qwen=dict(vocab_size=96, hidden_size=32, intermediate_size=128,
              num_hidden_layers=2, num_attention_heads=4, seq_length=32,
              max_position_embeddings=64),
    codegen=dict(vocab_size=96, n_embd=32, n_layer=2, n_head=4,
                 rotary_dim=4, max_position_embeddings=64),
    yuan=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
              num_hidden_layers=2, num_attention_heads=4,
              num_key_value_heads=2, max_position_embeddings=64),
    unified_transformer=dict(vocab_size=96, hidden_size=32,
                             intermediate_size=64, num_hidden_layers=2,
                             num_attention_heads=4,
                             max_position_embeddings=64),
    unimo=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
               num_hidden_layers=2, num_attention_heads=4,
               max_position_embeddings=64),
    chatglm=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                 num_attention_heads=4, inner_hidden_size=64,
                 max_sequence_length=64),
    jamba=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
               num_hidden_layers=4, num_attention_heads=4,
               num_key_value_heads=2, attn_layer_period=4,
               attn_layer_offset=1, expert_layer_period=2,
               expert_layer_offset=0, num_experts=4, num_experts_per_tok=2,
               mamba_d_state=8, max_position_embeddings=64),
    ctrl=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
              num_hidden_layers=2, num_attention_heads=4,
              resid_pdrop=0.0, embd_pdrop=0.0),
    reformer=dict(vocab_size=96, hidden_size=32, num_attention_heads=2,
                  attention_head_size=16, feed_forward_size=64,
                  attn_layers=("local", "lsh"), lsh_attn_chunk_length=8,
                  local_attn_chunk_length=8, num_hashes=2, num_buckets=4,
                  axial_pos_shape=(4, 8), hidden_dropout_prob=0.0),
    glm=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
             num_attention_heads=4, max_position_embeddings=64,
             hidden_dropout_prob=0.0),
    artist=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                num_hidden_layers=2, num_attention_heads=4,
                max_position_embeddings=64),
)
# gpt2 aliases gpt; skip the alias row
FAMILIES = sorted(mt for mt, entry in MODEL_REGISTRY.items()
                  if "causal_lm" in entry and mt != "gpt2")


@pytest.mark.parametrize("model_type", FAMILIES)
def test_causal_family_contract(model_type):
    assert model_type in TINY, f"add a tiny config for {model_type}"
    torch.manual_seed(0)
    cfg_cls = get_class(model_type, "config")
    lm_cls = get_class(model_type, "causal_lm")
    model = lm_cls(cfg_cls(**TINY[model_type]))

    ids = torch.randint(3, 96, (2, 12))
    labels = ids.clone()
    labels[:, :-1] = ids[:, 1:]
    labels[:, -1] = -100
    loss, logits = model(input_ids=ids, labels=labels)
    assert logits.shape == (2, 12, 96)
    assert 1.0 < float(loss) < 15.0, float(loss)
    loss.backward()
    grads = [p.grad for p in model.parameters() if p.requires_grad]
    assert any(g is not None and g.abs().sum() > 0 for g in grads)

    model.zero_grad(set_to_none=True)
    model.eval()
    with torch.no_grad():
        full = model(input_ids=ids)
        _, past = model(input_ids=ids[:, :-1], use_cache=True)
        step, _ = model(input_ids=ids[:, -1:], use_cache=True,
                        past_key_values=past)
    torch.testing.assert_close(step[:, 0], full[:, -1], rtol=2e-4, atol=2e-4)

    out, _ = model.generate(ids[:, :4], max_new_tokens=3, do_sample=False)
    assert out.shape == (2, 3)
