"""MBart + shared seq2seq family checks (pegasus has its own file).

Reference behavior: paddlenlp/transformers/mbart/modeling.py.
"""
import torch

from paddlenlp_amd.transformers import (
    AutoModelForSeq2SeqLM,
    MBartConfig,
    MBartForConditionalGeneration,
)

torch.manual_seed(0)

TINY = dict(vocab_size=100, d_model=32, encoder_layers=2, decoder_layers=2,
            encoder_attention_heads=4, decoder_attention_heads=4,
            encoder_ffn_dim=64, decoder_ffn_dim=64,
            max_position_embeddings=64)


def _model():
    return MBartForConditionalGeneration(MBartConfig(**TINY)).eval()


def test_mbart_forward_backward_tied():
    m = _model().train()
    src = torch.randint(3, 100, (2, 9))
    labels = torch.randint(3, 100, (2, 6))
    loss, logits = m(input_ids=src, labels=labels)
    assert logits.shape == (2, 6, 100)
    loss.backward()
    assert m.mbart.shared.weight.grad is not None
    assert m.lm_head.weight.data_ptr() == m.mbart.shared.weight.data_ptr()


def test_mbart_position_offset():
    # BART-family learned positions live at index +2
    from paddlenlp_amd.transformers.mbart.modeling import (
        MBartLearnedPositionalEmbedding,
    )
    emb = MBartLearnedPositionalEmbedding(8, 4)
    assert emb.weight.shape == (10, 4)
    torch.testing.assert_close(emb(0, 2), emb.weight[2:4])
    torch.testing.assert_close(emb(3, 1), emb.weight[5:6])


def test_mbart_cached_decode_matches_full():
    m = _model()
    src = torch.randint(3, 100, (2, 8))
    tgt = torch.randint(3, 100, (2, 5))
    with torch.no_grad():
        full = m(input_ids=src, decoder_input_ids=tgt)
        enc = m.mbart.encoder(src)
        past = None
        for t in range(tgt.shape[1]):
            logits, past, _ = m(decoder_input_ids=tgt[:, t:t + 1],
                                encoder_output=enc, past_key_values=past,
                                use_cache=True)
    torch.testing.assert_close(full[:, -1], logits[:, 0], rtol=1e-4, atol=1e-4)


def test_mbart_generate_and_roundtrip(tmp_path):
    m = _model()
    src = torch.randint(3, 100, (2, 7))
    out, _ = m.generate(src, max_new_tokens=5)
    assert out.shape[0] == 2
    outb, _ = m.generate(src, max_new_tokens=5, num_beams=3)
    assert outb.shape[0] == 2

    m.save_pretrained(str(tmp_path))
    m2 = AutoModelForSeq2SeqLM.from_pretrained(str(tmp_path))
    assert isinstance(m2, MBartForConditionalGeneration)
    tgt = torch.randint(3, 100, (2, 4))
    with torch.no_grad():
        torch.testing.assert_close(m(input_ids=src, decoder_input_ids=tgt),
                                   m2(input_ids=src, decoder_input_ids=tgt))


def test_blenderbot_pre_and_post_ln():
    from paddlenlp_amd.transformers import (
        BlenderbotConfig,
        BlenderbotForConditionalGeneration,
        BlenderbotSmallConfig,
        BlenderbotSmallForConditionalGeneration,
    )

    tiny = dict(vocab_size=100, d_model=32, encoder_layers=2,
                decoder_layers=2, encoder_attention_heads=4,
                decoder_attention_heads=4, encoder_ffn_dim=64,
                decoder_ffn_dim=64, max_position_embeddings=64)
    big = BlenderbotForConditionalGeneration(BlenderbotConfig(**tiny)).eval()
    small = BlenderbotSmallForConditionalGeneration(
        BlenderbotSmallConfig(**tiny)).eval()
    # pre-LN variant has the final stack layernorm; post-LN doesn't
    assert big.base.encoder.layer_norm is not None
    assert small.base.encoder.layer_norm is None
    src = torch.randint(3, 100, (2, 8))
    labels = torch.randint(3, 100, (2, 5))
    for m in (big, small):
        loss, logits = m(input_ids=src, labels=labels)
        assert logits.shape == (2, 5, 100)
        loss.backward()
        # cached decode parity
        tgt = torch.randint(3, 100, (2, 5))
        with torch.no_grad():
            full = m(input_ids=src, decoder_input_ids=tgt)
            enc = m.base.encoder(src)
            past = None
            for t in range(5):
                lg, past, _ = m(decoder_input_ids=tgt[:, t:t + 1],
                                encoder_output=enc, past_key_values=past,
                                use_cache=True)
        torch.testing.assert_close(full[:, -1], lg[:, 0],
                                   rtol=1e-4, atol=1e-4)
        out, _ = m.generate(src, max_new_tokens=4, num_beams=2)
        assert out.shape[0] == 2


def test_blenderbot_roundtrip(tmp_path):
    from paddlenlp_amd.transformers import (
        AutoModelForSeq2SeqLM,
        BlenderbotSmallConfig,
        BlenderbotSmallForConditionalGeneration,
    )

    m = BlenderbotSmallForConditionalGeneration(BlenderbotSmallConfig(
        vocab_size=100, d_model=32, encoder_layers=2, decoder_layers=2,
        encoder_attention_heads=4, decoder_attention_heads=4,
        encoder_ffn_dim=64, decoder_ffn_dim=64,
        max_position_embeddings=64)).eval()
    m.save_pretrained(str(tmp_path))
    m2 = AutoModelForSeq2SeqLM.from_pretrained(str(tmp_path))
    assert isinstance(m2, BlenderbotSmallForConditionalGeneration)
    src = torch.randint(3, 100, (1, 6))
    tgt = torch.randint(3, 100, (1, 4))
    with torch.no_grad():
        torch.testing.assert_close(m(input_ids=src, decoder_input_ids=tgt),
                                   m2(input_ids=src, decoder_input_ids=tgt))
