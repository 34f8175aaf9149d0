"""CRF layer, student families (tinybert/ppminilm/skep) and MiniLM
distillation utilities.

Reference behavior: paddlenlp/layers/crf.py,
paddlenlp/transformers/{tinybert,ppminilm,skep}/modeling.py,
paddlenlp/transformers/distill_utils.py.
"""
import itertools

import torch
import torch.nn as nn

from paddlenlp_amd.layers import LinearChainCrf, LinearChainCrfLoss, ViterbiDecoder
from paddlenlp_amd.transformers import (
    PPMiniLMConfig,
    PPMiniLMForSequenceClassification,
    SkepConfig,
    SkepCrfForTokenClassification,
    SkepForSequenceClassification,
    TinyBertConfig,
    TinyBertForPretraining,
    TinyBertForSequenceClassification,
    calc_minilm_loss,
    calc_multi_relation_loss,
    to_distill,
)
from paddlenlp_amd.transformers.bert import BertModel
from paddlenlp_amd.transformers.bert.configuration import BertConfig
from paddlenlp_amd.transformers.tinybert import TinyBertModel

torch.manual_seed(0)

TINY = dict(vocab_size=100, hidden_size=32, num_hidden_layers=2,
            num_attention_heads=4, intermediate_size=64,
            max_position_embeddings=64)


def test_crf_matches_brute_force():
    crf = LinearChainCrf(3)
    S = 4
    emis = torch.randn(1, S, 3)
    lengths = torch.tensor([S])
    # enumerate all 3^4 paths
    best, best_path, tot = -1e9, None, []
    t = crf.transitions
    with torch.no_grad():
        for path in itertools.product(range(3), repeat=S):
            s = t[crf.start_idx, path[0]] + emis[0, 0, path[0]]
            for i in range(1, S):
                s = s + t[path[i - 1], path[i]] + emis[0, i, path[i]]
            s = s + t[path[-1], crf.stop_idx]
            tot.append(s)
            if s > best:
                best, best_path = s, path
        logZ = torch.logsumexp(torch.stack(tot), 0)
        torch.testing.assert_close(crf(emis, lengths)[0], logZ,
                                   rtol=1e-5, atol=1e-5)
        score, path = ViterbiDecoder(crf.transitions)(emis, lengths)
        torch.testing.assert_close(score[0], best, rtol=1e-5, atol=1e-5)
        assert tuple(path[0].tolist()) == best_path


def test_crf_loss_nonnegative_and_trains():
    crf = LinearChainCrf(4)
    loss_fn = LinearChainCrfLoss(crf)
    emis = torch.randn(3, 6, 4, requires_grad=True)
    lengths = torch.tensor([6, 4, 5])
    labels = torch.randint(0, 4, (3, 6))
    loss = loss_fn(emis, lengths, labels)
    assert loss.item() >= 0  # logZ >= gold path score
    loss.backward()
    assert emis.grad is not None and crf.transitions.grad is not None


def test_tinybert_fit_dense_projects_to_teacher_width():
    m = TinyBertForPretraining(TinyBertConfig(fit_size=48, **TINY))
    ids = torch.randint(1, 100, (2, 10))
    outs = m(ids)
    assert len(outs) == TINY["num_hidden_layers"] + 1
    assert all(o.shape == (2, 10, 48) for o in outs)


def test_student_families_classify():
    ids = torch.randint(1, 100, (2, 10))
    labels = torch.tensor([0, 1])
    for cls, cfg in [
        (TinyBertForSequenceClassification, TinyBertConfig(**TINY)),
        (PPMiniLMForSequenceClassification, PPMiniLMConfig(**TINY)),
        (SkepForSequenceClassification, SkepConfig(**TINY)),
    ]:
        loss, logits = cls(cfg)(ids, labels=labels)
        assert logits.shape == (2, 2)
        loss.backward()


def test_skep_crf_tagger():
    m = SkepCrfForTokenClassification(SkepConfig(num_labels=5, **TINY))
    ids = torch.randint(1, 100, (2, 10))
    lens = torch.tensor([10, 7])
    labels = torch.randint(0, 5, (2, 10))
    loss = m(ids, seq_lens=lens, labels=labels)
    assert loss.item() >= 0
    loss.backward()
    paths = m.eval()(ids, seq_lens=lens)
    assert paths.shape == (2, 10)
    assert (paths[1, 7:] == 0).all()  # masked beyond length
    assert paths.max() < 5


def test_to_distill_and_minilm_loss():
    teacher = to_distill(BertModel(BertConfig(**TINY)))
    student = to_distill(TinyBertModel(TinyBertConfig(
        hidden_size=16, fit_size=32, vocab_size=100, num_hidden_layers=2,
        num_attention_heads=4, intermediate_size=32,
        max_position_embeddings=64)))
    ids = torch.randint(1, 100, (2, 10))
    teacher(ids)
    student(ids)
    (tq, tk, tv), = teacher.distill_qkv
    (sq, sk, sv), = student.distill_qkv
    assert tq.shape == (2, 4, 10, 8)   # teacher head_dim 8
    assert sq.shape == (2, 4, 10, 4)   # student head_dim 4
    kl = nn.KLDivLoss(reduction="batchmean")
    loss = sum(calc_minilm_loss(kl, s, t, num_relation_heads=4)
               for s, t in [(sq, tq), (sk, tk), (sv, tv)])
    loss.backward()
    assert loss.item() > 0
    # identical tensors -> zero relation loss
    z = calc_minilm_loss(kl, tq.detach(), tq.detach())
    assert abs(z.item()) < 1e-6


def test_multi_relation_loss_weights():
    kl = nn.KLDivLoss(reduction="batchmean")
    s = torch.randn(2, 4, 6, 4)
    t = torch.randn(2, 4, 6, 4)
    base = calc_multi_relation_loss(kl, s, t)
    tt_only = calc_minilm_loss(kl, s, t)
    torch.testing.assert_close(base, tt_only)
    mixed = calc_multi_relation_loss(kl, s, t, alpha=0.3, beta=0.3)
    assert mixed.item() > 0
