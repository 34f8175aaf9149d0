"""AutoNLP: automatic model selection for text classification.

Reference behavior: paddlenlp/experimental/autonlp/
(AutoTrainerBase:38 + AutoTrainerForTextClassification:52) — run a set of
candidate model/hyperparameter configs, track per-trial metrics, expose
evaluate/predict/export/to_taskflow over the best trial.  The reference
orchestrates trials with ray; here trials run sequentially through the
framework Trainer (deterministic, no external scheduler dependency).
"""
from __future__ import annotations

import json
import os
import shutil
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional

import torch

from ...trainer import Trainer
from ...trainer.training_args import TrainingArguments
from ...utils.log import logger


@dataclass
class TrialResult:
    trial_id: str
    config: Dict[str, Any]
    metrics: Dict[str, float]
    output_dir: str


class AutoTrainerForTextClassification:
    """train() runs every candidate; best trial selected by `greater_is_better`
    on `metric_for_best_model` (default eval_accuracy)."""

    def __init__(self, train_dataset, eval_dataset,
                 label_column: str = "labels",
                 text_column: str = "input_ids",
                 model_builder: Optional[Callable[[Dict[str, Any]], torch.nn.Module]] = None,
                 candidates: Optional[List[Dict[str, Any]]] = None,
                 metric_for_best_model: str = "eval_accuracy",
                 greater_is_better: bool = True,
                 output_dir: str = "autonlp_output",
                 data_collator=None,
                 compute_metrics: Optional[Callable] = None):
        self.train_dataset = train_dataset
        self.eval_dataset = eval_dataset
        self.label_column = label_column
        self.text_column = text_column
        self.model_builder = model_builder
        self.candidates = candidates or self._default_candidates()
        self.metric_for_best_model = metric_for_best_model
        self.greater_is_better = greater_is_better
        self.output_dir = output_dir
        self.data_collator = data_collator
        self.compute_metrics = compute_metrics or self._accuracy_metrics
        self.trial_results: List[TrialResult] = []

    @staticmethod
    def _default_candidates() -> List[Dict[str, Any]]:
        """Reference _model_candidates: a small LR x epochs grid."""
        return [
            {"trial_id": "lr3e-5", "learning_rate": 3e-5, "num_train_epochs": 1},
            {"trial_id": "lr5e-5", "learning_rate": 5e-5, "num_train_epochs": 1},
        ]

    @staticmethod
    def _accuracy_metrics(eval_preds) -> Dict[str, float]:
        logits, labels = eval_preds
        pred = torch.as_tensor(logits).argmax(-1)
        labels = torch.as_tensor(labels)
        return {"accuracy": float((pred == labels.view(pred.shape)).float().mean())}

    # ------------------------------------------------------------------
    def _construct_trainer(self, cand: Dict[str, Any]) -> Trainer:
        assert self.model_builder is not None, \
            "pass model_builder=lambda cand: model (no downloads offline)"
        model = self.model_builder(cand)
        args = TrainingArguments(
            output_dir=os.path.join(self.output_dir, cand["trial_id"]),
            do_train=True, do_eval=True,
            learning_rate=cand.get("learning_rate", 5e-5),
            num_train_epochs=cand.get("num_train_epochs", 1),
            per_device_train_batch_size=cand.get("batch_size", 8),
            per_device_eval_batch_size=cand.get("batch_size", 8),
            logging_steps=1000, save_steps=1 << 30, report_to=[],
        )
        return Trainer(
            model=model, args=args,
            train_dataset=self.train_dataset, eval_dataset=self.eval_dataset,
            data_collator=self.data_collator,
            compute_metrics=self.compute_metrics,
        )

    def train(self) -> List[TrialResult]:
        for cand in self.candidates:
            trial_id = cand.get("trial_id") or f"trial{len(self.trial_results)}"
            logger.info(f"AutoNLP trial {trial_id}: {cand}")
            trainer = self._construct_trainer(cand)
            trainer.train()
            metrics = trainer.evaluate()
            # quality metrics (accuracy etc.) come from the predict pass
            _, _, pred_metrics = trainer.predict(self.eval_dataset)
            metrics.update({f"eval_{k}": v for k, v in pred_metrics.items()})
            out = os.path.join(self.output_dir, trial_id)
            trainer.model.save_pretrained(out) if hasattr(
                trainer.model, "save_pretrained") else None
            with open(os.path.join(out, "trial.json"), "w") as f:
                json.dump({"config": {k: v for k, v in cand.items()},
                           "metrics": metrics}, f, indent=2, default=str)
            self.trial_results.append(
                TrialResult(trial_id, cand, metrics, out))
        return self.trial_results

    # ------------------------------------------------------------------
    def _best(self) -> TrialResult:
        assert self.trial_results, "call train() first"
        key = lambda t: t.metrics.get(self.metric_for_best_model, float("-inf"))
        return (max if self.greater_is_better else min)(self.trial_results, key=key)

    @property
    def best_trial(self) -> TrialResult:
        return self._best()

    def evaluate(self, trial_id: Optional[str] = None) -> Dict[str, float]:
        t = self._trial(trial_id)
        return t.metrics

    def _trial(self, trial_id: Optional[str]) -> TrialResult:
        if trial_id is None:
            return self._best()
        for t in self.trial_results:
            if t.trial_id == trial_id:
                return t
        raise ValueError(f"unknown trial {trial_id!r}")

    def export(self, export_path: str, trial_id: Optional[str] = None) -> str:
        """Copy the chosen trial's saved model to export_path (reference
        export:179)."""
        t = self._trial(trial_id)
        os.makedirs(export_path, exist_ok=True)
        for fname in os.listdir(t.output_dir):
            src = os.path.join(t.output_dir, fname)
            if os.path.isfile(src):
                shutil.copy2(src, os.path.join(export_path, fname))
        logger.info(f"AutoNLP exported trial {t.trial_id} -> {export_path}")
        return export_path

    def to_taskflow(self, trial_id: Optional[str] = None, **kwargs):
        """Wrap the chosen trial as a Taskflow('text_classification')
        (needs a tokenizer.json alongside the exported model)."""
        from ...taskflow import Taskflow

        t = self._trial(trial_id)
        return Taskflow("text_classification", model=t.output_dir, **kwargs)
