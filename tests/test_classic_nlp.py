"""Classic-NLP components: seq2vec encoders, TokenEmbedding, dataaug.

Reference behavior: paddlenlp/{seq2vec/encoder.py, embeddings/
token_embedding.py, dataaug/}.
"""
import numpy as np
import torch

from paddlenlp_amd.dataaug import (
    CharDelete,
    CharInsert,
    CharSubstitute,
    CharSwap,
    WordDelete,
    WordInsert,
    WordSubstitute,
    WordSwap,
)
from paddlenlp_amd.embeddings import TokenEmbedding
from paddlenlp_amd.seq2vec import (
    BoWEncoder,
    CNNEncoder,
    GRUEncoder,
    LSTMEncoder,
    RNNEncoder,
    TCNEncoder,
)

torch.manual_seed(0)


def test_seq2vec_encoders_shapes():
    B, S, E = 3, 12, 16
    x = torch.randn(B, S, E)
    mask = torch.ones(B, S)
    mask[:, 8:] = 0

    bow = BoWEncoder(E)
    assert bow(x, mask).shape == (B, E)
    # masked positions don't contribute
    x2 = x.clone()
    x2[:, 8:] = 99.0
    torch.testing.assert_close(bow(x, mask), bow(x2, mask))

    cnn = CNNEncoder(E, num_filter=8, ngram_filter_sizes=(2, 3))
    assert cnn(x).shape == (B, 16)
    cnn_p = CNNEncoder(E, num_filter=8, ngram_filter_sizes=(2, 3), output_dim=5)
    assert cnn_p(x).shape == (B, 5)
    assert cnn_p.get_output_dim() == 5

    for cls in (RNNEncoder, GRUEncoder, LSTMEncoder):
        enc = cls(E, 10)
        assert enc(x).shape == (B, 10)
        bi = cls(E, 10, direction="bidirect", pooling_type="max")
        assert bi(x, mask=mask).shape == (B, 20)
        assert bi.get_output_dim() == 20

    tcn = TCNEncoder(E, num_channels=[8, 8])
    assert tcn(x).shape == (B, 8)


def test_seq2vec_classifier_trains():
    emb = torch.nn.Embedding(50, 16)
    enc = LSTMEncoder(16, 12, pooling_type="mean")
    head = torch.nn.Linear(enc.get_output_dim(), 2)
    ids = torch.randint(0, 50, (4, 10))
    labels = torch.randint(0, 2, (4,))
    logits = head(enc(emb(ids)))
    loss = torch.nn.functional.cross_entropy(logits, labels)
    loss.backward()
    assert emb.weight.grad is not None


def test_token_embedding(tmp_path):
    emb = TokenEmbedding(vocab=["hello", "world", "foo"], embedding_dim=8)
    assert emb.vocab_size == 5  # + [PAD], [UNK]
    assert emb.get_idx_from_word("missing") == emb.unk_idx
    vecs = emb.search(["hello", "world"])
    assert vecs.shape == (2, 8)
    sim = emb.cosine_sim("hello", "hello")
    assert abs(sim - 1.0) < 1e-5

    # load from a text table
    p = tmp_path / "vecs.txt"
    p.write_text("2 3\naa 1 0 0\nbb 0 1 0\n")
    emb2 = TokenEmbedding(embedding_source=str(p))
    np.testing.assert_allclose(emb2.search("aa")[0], [1, 0, 0])
    assert emb2.cosine_sim("aa", "bb") < 1e-6


def test_word_augment():
    sub = WordSubstitute("custom", custom_dict={"cat": ["dog"]},
                         aug_n=1, create_n=2, seed=0)
    outs = sub.augment("the cat sat")
    assert len(outs) == 2 and all(o == "the dog sat" for o in outs)

    rnd = WordSubstitute("random", vocab=["x", "y"], aug_n=1, seed=0)
    out = rnd.augment("a b c")[0]
    assert out != "a b c" and len(out.split()) == 3

    ins = WordInsert(vocab=["NEW"], aug_n=1, seed=0)
    out = ins.augment("a b c")[0]
    assert len(out.split()) == 4 and "NEW" in out

    swap = WordSwap(aug_n=1, seed=0)
    out = swap.augment("a b c d")[0]
    assert sorted(out.split()) == ["a", "b", "c", "d"] and out != "a b c d"

    dele = WordDelete(aug_n=1, seed=0)
    out = dele.augment("a b c d")[0]
    assert len(out.split()) == 3

    # stop words are never touched
    sub2 = WordSubstitute("random", vocab=["z"], aug_n=5, aug_max=10,
                          stop_words=["keep"], seed=1)
    out = sub2.augment("keep keep keep")[0]
    assert out == "keep keep keep"


def test_char_augment():
    s = "hello world"
    for cls in (CharSubstitute, CharInsert, CharSwap, CharDelete):
        out = cls(aug_n=1, seed=3).augment(s)[0]
        assert out != s or cls is CharSwap  # swap can no-op on equal chars
        assert len(out.split()) == 2


def test_extended_metrics():
    from paddlenlp_amd.metrics import (
        MRR,
        Distinct,
        Mcc,
        PearsonAndSpearman,
        Rouge2,
        SpanEvaluator,
    )

    r2 = Rouge2()
    r2.update("the cat sat on".split(), ["the cat sat down".split()])
    assert 0 < r2.accumulate() < 1  # "the cat", "cat sat" match of 3

    d = Distinct(2)
    d.update(["a", "b", "a", "b"])
    assert d.accumulate() == 2 / 3  # (a,b) (b,a) (a,b) -> 2 unique of 3

    mcc = Mcc()
    mcc.update([1, 1, 0, 0], [1, 1, 0, 0])
    assert abs(mcc.accumulate() - 1.0) < 1e-9
    mcc.reset()
    mcc.update([1, 0], [0, 1])
    assert mcc.accumulate() <= 0

    ps = PearsonAndSpearman()
    ps.update([1.0, 2.0, 3.0], [2.0, 4.0, 6.0])
    out = ps.accumulate()
    assert abs(out["pearson"] - 1.0) < 1e-9
    assert abs(out["spearman"] - 1.0) < 1e-9

    mrr = MRR()
    mrr.update(1)
    mrr.update(2)
    mrr.update(0)
    assert abs(mrr.accumulate() - (1 + 0.5 + 0) / 3) < 1e-9

    se = SpanEvaluator()
    se.update([(0, 2), (5, 7)], [(0, 2), (3, 4)])
    p, r, f1 = se.accumulate()
    assert p == 0.5 and r == 0.5 and abs(f1 - 0.5) < 1e-9


def test_sighan_detection_correction_f1():
    from paddlenlp_amd.metrics import CorrectionF1, DetectionF1

    det = DetectionF1()
    det.update([2, 5], [2, 5])   # exact -> tp
    det.update([1], [1, 3])      # partial -> fp + fn
    det.update([], [])           # clean sentence, clean prediction
    p, r, f1 = det.accumulate()
    assert p == 0.5 and r == 0.5

    cor = CorrectionF1()
    cor.update([(2, "cat")], [(2, "cat")])
    cor.update([(2, "dog")], [(2, "cat")])  # right position, wrong token
    p, r, f1 = cor.accumulate()
    assert p == 0.5 and r == 0.5
