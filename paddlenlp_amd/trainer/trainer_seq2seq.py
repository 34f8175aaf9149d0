"""Seq2SeqTrainer: generation-based evaluation.

Reference: paddlenlp/trainer/trainer_seq2seq.py — evaluate/predict run
model.generate and hand decoded outputs to compute_metrics.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch

from ..generation import GenerationConfig
from .trainer import Trainer


class Seq2SeqTrainer(Trainer):
    def __init__(self, *args, gen_config: Optional[GenerationConfig] = None, **kwargs):
        super().__init__(*args, **kwargs)
        self.gen_config = gen_config or GenerationConfig(max_new_tokens=32)

    @torch.no_grad()
    def evaluate(self, eval_dataset=None, metric_key_prefix: str = "eval") -> Dict[str, float]:
        model = self._model_wrapped or self.model
        model.eval()
        dataloader = self.get_eval_dataloader(eval_dataset)
        preds, refs = [], []
        for i, inputs in enumerate(dataloader):
            if 0 < self.args.max_evaluate_steps <= i:
                break
            inputs = self._prepare_inputs(inputs)
            out, _ = model.generate(inputs["input_ids"], self.gen_config)
            preds.extend(out.cpu().tolist())
            if "labels" in inputs:
                refs.extend(inputs["labels"].cpu().tolist())
        model.train()
        metrics = {}
        if self.compute_metrics is not None:
            metrics = self.compute_metrics((preds, refs))
        metrics = {f"{metric_key_prefix}_{k}" if not k.startswith(metric_key_prefix) else k: v
                   for k, v in metrics.items()}
        self.log(dict(metrics))
        return metrics
