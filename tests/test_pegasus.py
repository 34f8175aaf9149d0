"""Pegasus seq2seq family (reference: paddlenlp/transformers/pegasus/)."""
import torch

from paddlenlp_amd.transformers import (
    AutoConfig,
    AutoModelForSeq2SeqLM,
    PegasusConfig,
    PegasusForConditionalGeneration,
    PegasusModel,
)

torch.manual_seed(0)

TINY = dict(vocab_size=100, d_model=32, encoder_layers=2, decoder_layers=2,
            encoder_attention_heads=4, decoder_attention_heads=4,
            encoder_ffn_dim=64, decoder_ffn_dim=64,
            max_position_embeddings=64)


def _model():
    return PegasusForConditionalGeneration(PegasusConfig(**TINY)).eval()


def test_pegasus_forward_backward():
    m = _model().train()
    src = torch.randint(2, 100, (2, 9))
    labels = torch.randint(2, 100, (2, 6))
    loss, logits = m(input_ids=src, labels=labels)
    assert logits.shape == (2, 6, 100)
    loss.backward()
    assert m.pegasus.shared.weight.grad is not None
    # embeddings and lm_head are one tensor
    assert m.lm_head.weight.data_ptr() == m.pegasus.shared.weight.data_ptr()


def test_pegasus_cached_decode_matches_full():
    m = _model()
    src = torch.randint(2, 100, (2, 8))
    tgt = torch.randint(2, 100, (2, 5))
    with torch.no_grad():
        full = m(input_ids=src, decoder_input_ids=tgt)
        enc = m.pegasus.encoder(src)
        past = None
        steps = []
        for t in range(tgt.shape[1]):
            logits, past, _ = m(decoder_input_ids=tgt[:, t:t + 1],
                                encoder_output=enc, past_key_values=past,
                                use_cache=True)
            steps.append(logits[:, 0])
    torch.testing.assert_close(full[:, -1], steps[-1], rtol=1e-4, atol=1e-4)


def test_pegasus_generate_greedy_and_beam():
    m = _model()
    src = torch.randint(2, 100, (2, 7))
    out, _ = m.generate(src, max_new_tokens=5)
    assert out.shape[0] == 2 and out.shape[1] <= 5
    outb, scores = m.generate(src, max_new_tokens=5, num_beams=3)
    assert outb.shape[0] == 2


def test_pegasus_sinusoidal_positions_fixed():
    from paddlenlp_amd.transformers.pegasus.modeling import sinusoidal_positions
    tab = sinusoidal_positions(16, 8)
    assert tab.shape == (16, 8)
    torch.testing.assert_close(tab[0], torch.zeros(8).scatter(
        0, torch.arange(1, 8, 2), 1.0))  # sin(0)=0, cos(0)=1
    # position table is lazily materialized, never in the state_dict
    m = _model()
    assert all("pos" not in k.split(".") or "table" not in k
               for k in m.state_dict())
    assert not any(".pos.table" in k for k in m.state_dict())


def test_pegasus_save_load_roundtrip(tmp_path):
    m = _model()
    m.save_pretrained(str(tmp_path))
    m2 = PegasusForConditionalGeneration.from_pretrained(str(tmp_path))
    src = torch.randint(2, 100, (1, 6))
    tgt = torch.randint(2, 100, (1, 4))
    with torch.no_grad():
        torch.testing.assert_close(m(input_ids=src, decoder_input_ids=tgt),
                                   m2(input_ids=src, decoder_input_ids=tgt))
    assert m2.lm_head.weight.data_ptr() == m2.pegasus.shared.weight.data_ptr()


def test_pegasus_auto_classes(tmp_path):
    m = _model()
    m.save_pretrained(str(tmp_path))
    cfg = AutoConfig.from_pretrained(str(tmp_path))
    assert isinstance(cfg, PegasusConfig)
    m2 = AutoModelForSeq2SeqLM.from_pretrained(str(tmp_path))
    assert isinstance(m2, PegasusForConditionalGeneration)
