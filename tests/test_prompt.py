"""Prompt learning: templates, verbalizers, prompt classification model.

Reference behavior: paddlenlp/prompt/{template,verbalizer,prompt_model}.py.
"""
import torch

from paddlenlp_amd.prompt import (
    ManualTemplate,
    ManualVerbalizer,
    PromptModelForSequenceClassification,
    SoftTemplate,
    SoftVerbalizer,
)
from paddlenlp_amd.transformers import BertConfig, BertForMaskedLM
from paddlenlp_amd.transformers.tokenizer_utils import PretrainedTokenizer

torch.manual_seed(0)

VOCAB = ["[PAD]", "[UNK]", "[MASK]", "it", "was", "great", "bad", "movie",
         "the", "good", "terrible"]


def make_tokenizer():
    from tokenizers import Tokenizer, models, pre_tokenizers

    t = Tokenizer(models.WordLevel({w: i for i, w in enumerate(VOCAB)},
                                   unk_token="[UNK]"))
    t.pre_tokenizer = pre_tokenizers.WhitespaceSplit()
    return PretrainedTokenizer(tokenizer=t, pad_token="[PAD]", unk_token="[UNK]")


def tiny_mlm():
    return BertForMaskedLM(BertConfig(
        vocab_size=len(VOCAB), hidden_size=32, num_hidden_layers=2,
        num_attention_heads=4, intermediate_size=64,
        max_position_embeddings=64, hidden_dropout_prob=0.0,
        attention_probs_dropout_prob=0.0))


def test_template_render():
    tok = make_tokenizer()
    tpl = ManualTemplate(
        [{"text": "text_a"}, {"hard": "it was"}, {"mask": None}], tok)
    r = tpl.render({"text_a": "great movie"})
    assert r["mask_position"] == len(r["input_ids"]) - 1
    assert r["input_ids"][r["mask_position"]] == 2  # [MASK]
    assert r["soft_positions"] == []

    soft = SoftTemplate(
        [{"soft": 3}, {"text": "text_a"}, {"mask": None}], tok)
    r2 = soft.render({"text_a": "bad"})
    assert r2["soft_positions"] == [0, 1, 2]
    assert soft.num_soft_tokens == 3


def test_manual_verbalizer_logits():
    tok = make_tokenizer()
    verb = ManualVerbalizer(tok, {"neg": ["bad", "terrible"],
                                  "pos": ["great", "good"]})
    assert verb.labels == ["neg", "pos"]
    mask_logits = torch.zeros(1, len(VOCAB))
    mask_logits[0, VOCAB.index("great")] = 4.0
    mask_logits[0, VOCAB.index("good")] = 2.0
    out = verb.process_logits(mask_logits)
    assert out.shape == (1, 2)
    assert out[0, 1] > out[0, 0]  # "pos" words dominate


def test_prompt_model_trains():
    tok = make_tokenizer()
    tpl = ManualTemplate(
        [{"text": "text_a"}, {"hard": "it was"}, {"mask": None}], tok)
    verb = ManualVerbalizer(tok, {"neg": ["bad"], "pos": ["great"]})
    model = PromptModelForSequenceClassification(
        tiny_mlm(), tpl, verb, freeze_plm=False)
    examples = [{"text_a": "great movie"}, {"text_a": "terrible movie"}]
    labels = torch.tensor([1, 0])
    loss, logits = model(examples, labels=labels)
    assert logits.shape == (2, 2)
    loss.backward()
    preds = model.predict(examples)
    assert all(p in ("neg", "pos") for p in preds)


def test_soft_prompt_gets_gradients():
    tok = make_tokenizer()
    tpl = SoftTemplate(
        [{"soft": 2}, {"text": "text_a"}, {"mask": None}], tok)
    verb = ManualVerbalizer(tok, {"neg": ["bad"], "pos": ["great"]})
    model = PromptModelForSequenceClassification(
        tiny_mlm(), tpl, verb, freeze_plm=True)
    assert model.soft_embeddings is not None
    loss, _ = model([{"text_a": "movie"}], labels=torch.tensor([1]))
    loss.backward()
    assert model.soft_embeddings.grad is not None
    assert model.soft_embeddings.grad.abs().sum() > 0
    # frozen backbone got no grads
    assert all(p.grad is None for p in model.plm.parameters())


def test_soft_verbalizer_init():
    tok = make_tokenizer()
    mlm = tiny_mlm()
    head_w = mlm.get_input_embeddings().weight
    sv = SoftVerbalizer(tok, {"neg": ["bad"], "pos": ["great"]}, head_w)
    torch.testing.assert_close(sv.head.weight[1], head_w[VOCAB.index("great")])
    hidden = torch.randn(2, 32)
    assert sv.process_hidden(hidden).shape == (2, 2)
