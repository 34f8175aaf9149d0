"""Compression strategies: DynaBERT width pruning, PTQ calibration grid,
QAT fake-quant, embedding quantization.

Reference behavior: paddlenlp/trainer/trainer_compress.py + compression_args.py.
"""
import pytest
import torch

from paddlenlp_amd.trainer.compression_args import CompressionArguments
from paddlenlp_amd.trainer.trainer_compress import (
    A8W8Linear,
    QATLinear,
    QuantEmbedding,
    distill_step,
    dynabert_prune,
    fake_quant,
    post_training_quantization,
    quant_aware_training,
    quantize_embeddings,
)
from paddlenlp_amd.transformers import BertConfig, BertForSequenceClassification

torch.manual_seed(0)


def tiny_cls(num_labels=3):
    cfg = BertConfig(vocab_size=120, hidden_size=32, num_hidden_layers=2,
                     num_attention_heads=4, intermediate_size=64,
                     max_position_embeddings=64, hidden_dropout_prob=0.0,
                     attention_probs_dropout_prob=0.0, num_labels=num_labels)
    return BertForSequenceClassification(cfg)


def batches(n=6, B=4, S=12):
    out = []
    g = torch.Generator().manual_seed(1)
    for _ in range(n):
        out.append({
            "input_ids": torch.randint(0, 120, (B, S), generator=g),
            "labels": torch.randint(0, 3, (B,), generator=g),
        })
    return out


def test_compression_args_strategies():
    args = CompressionArguments(strategy="dynabert+ptq")
    assert args.strategies == ["dynabert", "ptq"]
    with pytest.raises(ValueError):
        CompressionArguments(strategy="bogus")


def test_ptq_quantizes_and_stays_close():
    model = tiny_cls().eval()
    data = batches()
    ids = data[0]["input_ids"]
    with torch.no_grad():
        ref = model(ids)
    for algo in ("abs_max", "avg", "mse"):
        import copy
        q = post_training_quantization(copy.deepcopy(model), iter(data), algo)
        n_q = sum(isinstance(m, A8W8Linear) for m in q.modules())
        assert n_q > 0
        # classifier head left fp
        assert isinstance(q.classifier, torch.nn.Linear)
        with torch.no_grad():
            out = q(ids)
        rel = (out - ref).abs().mean() / ref.abs().mean().clamp(min=1e-6)
        assert rel < 0.5, (algo, rel)  # int8 sim keeps logits in the ballpark


def test_qat_fake_quant_ste_grads():
    x = torch.randn(8, requires_grad=True)
    y = fake_quant(x).sum()
    y.backward()
    torch.testing.assert_close(x.grad, torch.ones_like(x))  # STE passthrough

    model = tiny_cls()
    quant_aware_training(model)
    assert any(isinstance(m, QATLinear) for m in model.modules())
    b = batches(1)[0]
    loss, _ = model(b["input_ids"], labels=b["labels"])
    loss.backward()
    qkv = model.bert.encoder.layers[0].self_attn.qkv_proj
    assert isinstance(qkv, QATLinear) and qkv.weight.grad is not None
    assert qkv.weight.grad.abs().sum() > 0


def test_dynabert_prune_and_distill():
    model = tiny_cls()
    data = batches()
    student = dynabert_prune(model, iter(data), width_mult=0.5, num_batches=2)
    # width halved: 4 heads -> 2, ffn 64 -> 32
    attn = student.bert.encoder.layers[0].self_attn
    assert attn.num_heads == 2
    assert attn.qkv_proj.out_features == 3 * 2 * attn.head_dim
    assert student.bert.encoder.layers[0].fc_in.out_features == 32
    n_student = sum(p.numel() for p in student.parameters())
    n_teacher = sum(p.numel() for p in model.parameters())
    assert n_student < n_teacher

    # pruned model still runs and distills
    b = data[0]
    loss = distill_step(student, model, b)
    loss.backward()
    assert torch.isfinite(loss)


def test_quantize_embeddings():
    model = tiny_cls()
    ids = torch.randint(0, 120, (2, 10))
    with torch.no_grad():
        ref = model(ids)
    quantize_embeddings(model)
    assert isinstance(model.bert.embeddings.word_embeddings, QuantEmbedding)
    with torch.no_grad():
        out = model(ids)
    rel = (out - ref).abs().mean() / ref.abs().mean().clamp(min=1e-6)
    assert rel < 0.5


def test_vector_observers_channel_and_headwise():
    """channel_wise / abs_max_headwise observers keep vector scales
    (reference llm/experimental/observer/)."""
    import torch

    from paddlenlp_amd.trainer.trainer_compress import _ActObserver

    obs = _ActObserver("channel_wise")
    obs.update(torch.tensor([[1.0, -2.0, 0.5, 4.0]]))
    obs.update(torch.tensor([[3.0, 1.0, 0.1, 0.2]]))
    assert torch.allclose(obs.scale(), torch.tensor([3.0, 2.0, 0.5, 4.0]))

    head = _ActObserver("abs_max_headwise", heads=2)
    head.update(torch.tensor([[1.0, -2.0, 0.5, 4.0]]))
    assert torch.allclose(head.scale(), torch.tensor([2.0, 4.0]))
