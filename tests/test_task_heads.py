"""Synthesized task heads: every registered family answers the four
AutoModelFor* task classes (reference parity for the head matrix)."""
import pytest
import torch

from paddlenlp_amd.transformers.auto.registry import MODEL_REGISTRY, get_class

# families whose base model is a plain encoder/decoder usable under a head
PROBE = ["nezha", "mpnet", "llama", "gpt2", "electra", "fnet"]


@pytest.mark.parametrize("model_type", PROBE)
@pytest.mark.parametrize("kind", ["sequence_classification",
                                  "token_classification",
                                  "question_answering",
                                  "multiple_choice"])
def test_head_synthesis_forward(model_type, kind):
    if model_type not in MODEL_REGISTRY:
        pytest.skip(f"{model_type} not registered")
    cfg_cls = get_class(model_type, "config")
    try:
        head_cls = get_class(model_type, kind)
    except ValueError:
        pytest.skip(f"{model_type} has no base model entry")
    cfg_kwargs = dict(vocab_size=64, hidden_size=32, num_hidden_layers=1,
                      num_attention_heads=2, intermediate_size=64,
                      max_position_embeddings=64, num_labels=3)
    cfg = None
    for drop in range(3):
        try:
            cfg = cfg_cls(**cfg_kwargs)
            break
        except TypeError as e:
            # configs differ slightly; drop the offending kwarg
            missing = str(e).split("'")[-2] if "'" in str(e) else None
            if missing and missing in cfg_kwargs:
                cfg_kwargs.pop(missing)
            else:
                raise
    model = head_cls.from_config(cfg) if hasattr(head_cls, "from_config") else head_cls(cfg)
    model.eval()
    B, S = 2, 8
    if kind == "multiple_choice":
        ids = torch.randint(3, 64, (B, 3, S))
        labels = torch.randint(0, 3, (B,))
        loss, logits = model(ids, labels=labels)
        assert logits.shape == (B, 3)
    elif kind == "question_answering":
        import inspect

        ids = torch.randint(3, 64, (B, S))
        start, end = model(ids)
        assert start.shape == (B, S) and end.shape == (B, S)
        params = inspect.signature(type(model).forward).parameters
        if "start_positions" in params:   # hand-written heads may omit loss
            loss, *_ = model(ids, start_positions=torch.tensor([1, 2]),
                             end_positions=torch.tensor([3, 4]))
        else:
            loss = start.sum() * 0
    else:
        ids = torch.randint(3, 64, (B, S))
        labels = (torch.randint(0, 3, (B,)) if kind == "sequence_classification"
                  else torch.randint(0, 3, (B, S)))
        loss, logits = model(ids, labels=labels)
    assert torch.isfinite(loss), (model_type, kind, loss)


def test_auto_multiple_choice_class_exists():
    from paddlenlp_amd.transformers.auto import AutoModelForMultipleChoice

    assert AutoModelForMultipleChoice._kind == "multiple_choice"
