"""Chinese-CLIP / BiT / ERNIE-Code / BertJapaneseTokenizer.

Reference behavior: paddlenlp/transformers/{chineseclip,bit,ernie_code,
bert_japanese}/.
"""
import torch

from paddlenlp_amd.transformers import (
    BertJapaneseTokenizer,
    BitConfig,
    BitForImageClassification,
    BitModel,
    ChineseCLIPConfig,
    ChineseCLIPModel,
    ErnieCodeConfig,
    ErnieCodeForConditionalGeneration,
)


def test_chineseclip_bert_text_tower_and_contrastive_logits():
    torch.manual_seed(0)
    cfg = ChineseCLIPConfig(
        text_config=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64),
        vision_config=dict(hidden_size=32, num_hidden_layers=2,
                           num_attention_heads=4, intermediate_size=64,
                           image_size=32, patch_size=8),
        projection_dim=16)
    m = ChineseCLIPModel(cfg).eval()
    ids = torch.randint(0, 96, (3, 10))
    px = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        lt, li = m(ids, px)
    assert lt.shape == (3, 2) and li.shape == (2, 3)
    # text tower is BERT-style: bidirectional — swapping a later token
    # changes the [CLS] feature (a causal CLIP tower would too, but the
    # reverse direction wouldn't; check future -> CLS flow)
    ids2 = ids.clone()
    ids2[:, -1] = (ids2[:, -1] + 1) % 96
    with torch.no_grad():
        f1 = m.get_text_features(ids)
        f2 = m.get_text_features(ids2)
    assert not torch.allclose(f1, f2, atol=1e-5)


def test_bit_weight_standardized_resnet():
    from paddlenlp_amd.transformers.bit.modeling import WSConv2d

    torch.manual_seed(0)
    cfg = BitConfig(embedding_size=16, hidden_sizes=(16, 32),
                    depths=(1, 1), num_groups=8, num_labels=5)
    m = BitForImageClassification(cfg)
    px = torch.randn(2, 3, 32, 32)
    loss, logits = m(px, labels=torch.tensor([0, 3]))
    assert logits.shape == (2, 5)
    loss.backward()
    # weight standardization: effective conv weights are zero-mean
    conv = m.bit.stages[0][0].conv2
    assert isinstance(conv, WSConv2d)
    w = conv.weight
    mean = w.mean(dim=(1, 2, 3), keepdim=True)
    var = w.var(dim=(1, 2, 3), keepdim=True, unbiased=False)
    ws = (w - mean) * torch.rsqrt(var + 1e-10)
    assert ws.mean(dim=(1, 2, 3)).abs().max() < 1e-5
    # no BatchNorm anywhere (BiT uses GroupNorm only)
    assert not any(isinstance(mod, torch.nn.BatchNorm2d)
                   for mod in m.modules())


def test_ernie_code_is_t5_shaped():
    torch.manual_seed(0)
    cfg = ErnieCodeConfig(vocab_size=96, d_model=32, d_kv=8, d_ff=64,
                          num_layers=2, num_heads=4)
    m = ErnieCodeForConditionalGeneration(cfg)
    src = torch.randint(0, 96, (2, 8))
    labels = torch.randint(0, 96, (2, 6))
    out = m(input_ids=src, labels=labels)
    loss = out[0]
    loss.backward()
    assert float(loss) > 0


def test_bert_japanese_tokenizer(tmp_path):
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "日", "本", "語", "hello", "world", "hel", "##lo"]
    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join(vocab), encoding="utf-8")
    # character subword mode: unsegmented Japanese -> per-char ids
    tok = BertJapaneseTokenizer(vocab_file=str(vf),
                                subword_tokenizer_type="character")
    ids = tok._tokenizer.encode("日本語").ids
    assert ids == [5, 6, 7]
    # wordpiece mode on spaced text
    tok2 = BertJapaneseTokenizer(vocab_file=str(vf),
                                 subword_tokenizer_type="wordpiece")
    ids2 = tok2._tokenizer.encode("hello world").ids
    assert ids2 == [8, 9]
    # mecab request degrades gracefully
    import warnings

    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        BertJapaneseTokenizer(vocab_file=str(vf),
                              word_tokenizer_type="mecab")
        assert any("MeCab" in str(x.message) for x in w)
