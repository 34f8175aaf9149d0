"""FNet / ERNIE-Gram / RemBERT / XLM encoder-tail families.

Reference behavior: paddlenlp/transformers/{fnet,ernie_gram,rembert,xlm}/.
"""
import torch

from paddlenlp_amd.transformers import (
    ErnieGramConfig,
    ErnieGramForQuestionAnswering,
    ErnieGramForSequenceClassification,
    FNetConfig,
    FNetForMaskedLM,
    FNetForSequenceClassification,
    FNetModel,
    RemBertConfig,
    RemBertForMaskedLM,
    RemBertModel,
    XLMConfig,
    XLMForSequenceClassification,
    XLMModel,
    XLMWithLMHeadModel,
)

torch.manual_seed(0)

TINY = dict(vocab_size=100, hidden_size=32, num_hidden_layers=2,
            num_attention_heads=4, intermediate_size=64,
            max_position_embeddings=64)


def test_fnet_is_attention_free():
    m = FNetModel(FNetConfig(**TINY))
    names = [n for n, _ in m.named_parameters()]
    assert not any("qkv" in n or "attn" in n for n in names)
    ids = torch.randint(4, 100, (2, 10))
    seq, pooled = m(ids)
    assert seq.shape == (2, 10, 32) and pooled.shape == (2, 32)


def test_fnet_fourier_mixing_moves_information():
    # FFT mixing is global: changing token 0 must change the layer-1 output
    # at EVERY position (an attention-free smoke test of the mixing)
    m = FNetModel(FNetConfig(hidden_dropout_prob=0.0, **TINY)).eval()
    ids = torch.randint(4, 100, (1, 8))
    ids2 = ids.clone()
    ids2[0, 0] = (ids[0, 0] + 1) % 96 + 4
    with torch.no_grad():
        a, _ = m(ids)
        b, _ = m(ids2)
    assert (a - b).abs().amax(dim=-1).min() > 0  # all positions changed


def test_fnet_heads():
    ids = torch.randint(4, 100, (2, 10))
    loss, logits = FNetForMaskedLM(FNetConfig(**TINY))(ids, labels=ids)
    loss.backward()
    assert logits.shape == (2, 10, 100)
    loss, _ = FNetForSequenceClassification(FNetConfig(**TINY))(
        ids, labels=torch.tensor([0, 1]))
    loss.backward()


def test_ernie_gram_heads():
    ids = torch.randint(2, 100, (2, 10))
    loss, logits = ErnieGramForSequenceClassification(ErnieGramConfig(**TINY))(
        ids, labels=torch.tensor([0, 1]))
    loss.backward()
    start, end = ErnieGramForQuestionAnswering(ErnieGramConfig(**TINY))(ids)
    assert start.shape == (2, 10)


def test_rembert_decoupled_embeddings():
    cfg = RemBertConfig(input_embedding_size=16, output_embedding_size=24,
                        **TINY)
    m = RemBertModel(cfg)
    assert m.word_embeddings.weight.shape == (100, 16)
    assert m.embedding_hidden_mapping_in.weight.shape == (32, 16)
    ids = torch.randint(2, 100, (2, 10))
    seq, _ = m(ids)
    assert seq.shape == (2, 10, 32)
    mlm = RemBertForMaskedLM(cfg)
    # untied: output path has its own 24-wide projection
    assert mlm.dense.weight.shape == (24, 32)
    assert mlm.decoder.weight.shape == (100, 24)
    loss, _ = mlm(ids, labels=ids)
    loss.backward()


def test_xlm_language_embeddings():
    cfg = XLMConfig(n_langs=3, hidden_dropout_prob=0.0,
                    attention_probs_dropout_prob=0.0, **TINY)
    m = XLMModel(cfg).eval()
    ids = torch.randint(2, 100, (1, 8))
    with torch.no_grad():
        a = m(ids, langs=torch.zeros(1, 8, dtype=torch.long))
        b = m(ids, langs=torch.ones(1, 8, dtype=torch.long))
    assert not torch.allclose(a, b)  # language embedding matters

    lm = XLMWithLMHeadModel(cfg)
    assert lm.lm_head.weight.data_ptr() == \
        lm.xlm.word_embeddings.weight.data_ptr()
    loss, _ = lm(ids, langs=torch.zeros(1, 8, dtype=torch.long), labels=ids)
    loss.backward()


def test_xlm_sinusoidal_positions():
    cfg = XLMConfig(use_sinusoidal_embeddings=True, **TINY)
    m = XLMForSequenceClassification(cfg)
    assert m.xlm.position_embeddings is None
    ids = torch.randint(2, 100, (2, 10))
    loss, logits = m(ids, labels=torch.tensor([0, 1]))
    loss.backward()
    assert logits.shape == (2, 2)
    # sinusoid table never serialized
    assert all("position_table" not in k for k in m.state_dict())


def test_megatronbert_pre_ln():
    from paddlenlp_amd.transformers import (
        MegatronBertConfig,
        MegatronBertForQuestionAnswering,
        MegatronBertForSequenceClassification,
        MegatronBertModel,
    )

    cfg = MegatronBertConfig(**TINY)
    m = MegatronBertModel(cfg)
    # pre-LN tower: extra final LayerNorm after the stack
    assert isinstance(m.ln, torch.nn.LayerNorm)
    ids = torch.randint(2, 100, (2, 10))
    seq, pooled = m(ids)
    assert seq.shape == (2, 10, 32)
    loss, _ = MegatronBertForSequenceClassification(cfg)(
        ids, labels=torch.tensor([0, 1]))
    loss.backward()
    start, end = MegatronBertForQuestionAnswering(cfg)(ids)
    assert start.shape == (2, 10)


def test_layoutlm_bbox_embeddings():
    from paddlenlp_amd.transformers import (
        LayoutLMConfig,
        LayoutLMForTokenClassification,
        LayoutLMModel,
    )

    cfg = LayoutLMConfig(max_2d_position_embeddings=128,
                         hidden_dropout_prob=0.0,
                         attention_probs_dropout_prob=0.0, **TINY)
    m = LayoutLMModel(cfg).eval()
    ids = torch.randint(2, 100, (2, 10))
    bbox = torch.randint(0, 100, (2, 10, 4))
    bbox[..., 2:] = torch.minimum(
        bbox[..., :2] + torch.randint(1, 20, (2, 10, 2)),
        torch.tensor(127))
    with torch.no_grad():
        a, _ = m(ids, bbox=bbox)
        b, _ = m(ids)  # no layout
        bbox2 = bbox.clone()
        # move one box's right edge (stays >= x0 and < 128)
        bbox2[0, 0, 2] = 127 if bbox[0, 0, 2] != 127 else 126
        c, _ = m(ids, bbox=bbox2)
    assert not torch.allclose(a, b)  # layout matters
    assert not torch.allclose(a[0], c[0])  # a single bbox matters
    loss, logits = LayoutLMForTokenClassification(cfg)(
        ids, bbox=bbox, labels=torch.randint(0, 2, (2, 10)))
    loss.backward()
    assert logits.shape == (2, 10, 2)
