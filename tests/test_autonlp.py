"""AutoNLP: candidate trials, best-model selection, export, taskflow handoff.

Reference behavior: paddlenlp/experimental/autonlp/text_classification.py.
"""
import os

import torch

from paddlenlp_amd.experimental.autonlp import AutoTrainerForTextClassification
from paddlenlp_amd.transformers import BertConfig, BertForSequenceClassification

torch.manual_seed(0)


class _ClsDataset(torch.utils.data.Dataset):
    """Separable toy task: label = token id bucket."""

    def __init__(self, n=32, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.items = []
        for _ in range(n):
            label = int(torch.randint(0, 2, (1,), generator=g))
            lo, hi = (3, 30) if label == 0 else (60, 110)
            ids = torch.randint(lo, hi, (12,), generator=g)
            self.items.append({"input_ids": ids, "labels": torch.tensor(label)})

    def __len__(self):
        return len(self.items)

    def __getitem__(self, i):
        return self.items[i]


def _builder(cand):
    torch.manual_seed(1)
    return BertForSequenceClassification(BertConfig(
        vocab_size=128, hidden_size=32, num_hidden_layers=2,
        num_attention_heads=4, intermediate_size=64,
        max_position_embeddings=64, num_labels=2,
        hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0))


def test_autonlp_trials_and_export(tmp_path):
    auto = AutoTrainerForTextClassification(
        train_dataset=_ClsDataset(48, seed=0),
        eval_dataset=_ClsDataset(16, seed=1),
        model_builder=_builder,
        candidates=[
            {"trial_id": "fast", "learning_rate": 5e-4, "num_train_epochs": 2},
            {"trial_id": "slow", "learning_rate": 1e-6, "num_train_epochs": 1},
        ],
        output_dir=str(tmp_path / "runs"),
    )
    results = auto.train()
    assert len(results) == 2
    assert all("eval_accuracy" in r.metrics for r in results)

    best = auto.best_trial
    # the trained (fast) trial must beat the barely-trained one
    assert best.trial_id == "fast", [r.metrics for r in results]
    assert best.metrics["eval_accuracy"] >= 0.7

    # per-trial evaluate + unknown-trial error
    assert auto.evaluate("slow") == results[1].metrics
    try:
        auto.evaluate("nope")
        raised = False
    except ValueError:
        raised = True
    assert raised

    export = auto.export(str(tmp_path / "best"))
    assert os.path.isfile(os.path.join(export, "config.json"))
    assert os.path.isfile(os.path.join(export, "trial.json"))

    m = BertForSequenceClassification.from_pretrained(export)
    assert m.config.num_labels == 2
