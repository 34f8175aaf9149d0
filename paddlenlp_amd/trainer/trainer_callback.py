"""Callback/event system.

Reference: paddlenlp/trainer/trainer_callback.py — TrainerState :47,
TrainerControl :118, TrainerCallback :167, CallbackHandler :301,
DefaultFlowCallback :432, ProgressCallback :478, EarlyStoppingCallback :540.
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.log import logger


@dataclass
class TrainerState:
    epoch: float = 0.0
    global_step: int = 0
    max_steps: int = 0
    num_train_epochs: int = 0
    log_history: List[Dict] = field(default_factory=list)
    best_metric: Optional[float] = None
    best_model_checkpoint: Optional[str] = None
    is_world_process_zero: bool = True
    trial_params: Optional[Dict] = None
    consumed_samples: int = 0

    def save_to_json(self, json_path: str):
        with open(json_path, "w", encoding="utf-8") as f:
            json.dump(dataclasses.asdict(self), f, indent=2, sort_keys=True)

    @classmethod
    def load_from_json(cls, json_path: str) -> "TrainerState":
        with open(json_path, "r", encoding="utf-8") as f:
            return cls(**json.load(f))


@dataclass
class TrainerControl:
    should_training_stop: bool = False
    should_epoch_stop: bool = False
    should_save: bool = False
    should_evaluate: bool = False
    should_log: bool = False

    def _new_training(self):
        self.should_training_stop = False

    def _new_epoch(self):
        self.should_epoch_stop = False

    def _new_step(self):
        self.should_save = False
        self.should_evaluate = False
        self.should_log = False


class TrainerCallback:
    def on_init_end(self, args, state, control, **kwargs):
        pass

    def on_train_begin(self, args, state, control, **kwargs):
        pass

    def on_train_end(self, args, state, control, **kwargs):
        pass

    def on_epoch_begin(self, args, state, control, **kwargs):
        pass

    def on_epoch_end(self, args, state, control, **kwargs):
        pass

    def on_step_begin(self, args, state, control, **kwargs):
        pass

    def on_step_end(self, args, state, control, **kwargs):
        pass

    def on_substep_end(self, args, state, control, **kwargs):
        pass

    def on_evaluate(self, args, state, control, metrics=None, **kwargs):
        pass

    def on_save(self, args, state, control, **kwargs):
        pass

    def on_log(self, args, state, control, logs=None, **kwargs):
        pass

    def on_prediction_step(self, args, state, control, **kwargs):
        pass


class CallbackHandler(TrainerCallback):
    def __init__(self, callbacks, model, tokenizer, optimizer, lr_scheduler):
        self.callbacks = []
        for cb in callbacks:
            self.add_callback(cb)
        self.model = model
        self.tokenizer = tokenizer
        self.optimizer = optimizer
        self.lr_scheduler = lr_scheduler
        self.train_dataloader = None

    def add_callback(self, callback):
        cb = callback() if isinstance(callback, type) else callback
        self.callbacks.append(cb)

    def pop_callback(self, callback):
        for cb in self.callbacks:
            if cb == callback or type(cb) == callback:
                self.callbacks.remove(cb)
                return cb

    def remove_callback(self, callback):
        self.pop_callback(callback)

    def call_event(self, event, args, state, control, **kwargs):
        for callback in self.callbacks:
            result = getattr(callback, event)(
                args, state, control,
                model=self.model, tokenizer=self.tokenizer,
                optimizer=self.optimizer, lr_scheduler=self.lr_scheduler,
                train_dataloader=self.train_dataloader, **kwargs,
            )
            if result is not None:
                control = result
        return control

    def on_train_begin(self, args, state, control, **kw):
        control._new_training()
        return self.call_event("on_train_begin", args, state, control, **kw)

    def on_train_end(self, args, state, control, **kw):
        return self.call_event("on_train_end", args, state, control, **kw)

    def on_epoch_begin(self, args, state, control, **kw):
        control._new_epoch()
        return self.call_event("on_epoch_begin", args, state, control, **kw)

    def on_epoch_end(self, args, state, control, **kw):
        return self.call_event("on_epoch_end", args, state, control, **kw)

    def on_step_begin(self, args, state, control, **kw):
        control._new_step()
        return self.call_event("on_step_begin", args, state, control, **kw)

    def on_step_end(self, args, state, control, **kw):
        return self.call_event("on_step_end", args, state, control, **kw)

    def on_substep_end(self, args, state, control, **kw):
        return self.call_event("on_substep_end", args, state, control, **kw)

    def on_evaluate(self, args, state, control, metrics=None, **kw):
        control.should_evaluate = False
        return self.call_event("on_evaluate", args, state, control, metrics=metrics, **kw)

    def on_save(self, args, state, control, **kw):
        control.should_save = False
        return self.call_event("on_save", args, state, control, **kw)

    def on_log(self, args, state, control, logs=None, **kw):
        control.should_log = False
        return self.call_event("on_log", args, state, control, logs=logs, **kw)

    def on_prediction_step(self, args, state, control, **kw):
        return self.call_event("on_prediction_step", args, state, control, **kw)


class DefaultFlowCallback(TrainerCallback):
    """Decides when to log / evaluate / save (reference :432)."""

    def on_step_end(self, args, state, control, **kwargs):
        if (getattr(args, "logging_strategy", "steps") == "steps"
                and args.logging_steps > 0
                and state.global_step % args.logging_steps == 0):
            control.should_log = True
        if (
            args.evaluation_strategy == "steps"
            and args.eval_steps
            and state.global_step % args.eval_steps == 0
        ):
            control.should_evaluate = True
        if (
            args.save_strategy == "steps"
            and args.save_steps > 0
            and state.global_step % args.save_steps == 0
        ):
            control.should_save = True
        if state.global_step >= state.max_steps:
            control.should_training_stop = True
        return control

    def on_epoch_end(self, args, state, control, **kwargs):
        if getattr(args, "logging_strategy", "steps") == "epoch":
            control.should_log = True
        if args.evaluation_strategy == "epoch":
            control.should_evaluate = True
        if args.save_strategy == "epoch":
            control.should_save = True
        return control


class ProgressCallback(TrainerCallback):
    def on_log(self, args, state, control, logs=None, **kwargs):
        if state.is_world_process_zero and logs:
            logger.info(f"step {state.global_step}: {json.dumps(logs, default=str)}")


class EarlyStoppingCallback(TrainerCallback):
    def __init__(self, early_stopping_patience: int = 1, early_stopping_threshold: float = 0.0):
        self.patience = early_stopping_patience
        self.threshold = early_stopping_threshold
        self.counter = 0

    def on_evaluate(self, args, state, control, metrics=None, **kwargs):
        metric = (metrics or {}).get("eval_loss")
        if metric is None:
            return
        if state.best_metric is None or metric < state.best_metric - self.threshold:
            state.best_metric = metric
            self.counter = 0
        else:
            self.counter += 1
            if self.counter >= self.patience:
                control.should_training_stop = True
