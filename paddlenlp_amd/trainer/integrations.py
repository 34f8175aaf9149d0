"""Logging-backend callbacks (reference: paddlenlp/trainer/integrations.py —
VisualDLCallback :78, TensorBoardCallback :162, WandbCallback :236)."""
from __future__ import annotations

import json
import os

from ..utils.log import logger
from .trainer_callback import TrainerCallback


def is_tensorboard_available() -> bool:
    try:

        return True
    except ImportError:
        return False


class TensorBoardCallback(TrainerCallback):
    def __init__(self, writer=None):
        self.writer = writer

    def _init_writer(self, args):
        if self.writer is None and is_tensorboard_available():
            from torch.utils.tensorboard import SummaryWriter

            os.makedirs(args.logging_dir, exist_ok=True)
            self.writer = SummaryWriter(log_dir=args.logging_dir)

    def on_train_begin(self, args, state, control, **kwargs):
        if state.is_world_process_zero:
            self._init_writer(args)
            if self.writer:
                self.writer.add_text("args", args.to_json_string())

    def on_log(self, args, state, control, logs=None, **kwargs):
        if not state.is_world_process_zero or self.writer is None:
            return
        for k, v in (logs or {}).items():
            if isinstance(v, (int, float)):
                self.writer.add_scalar(k, v, state.global_step)
        self.writer.flush()

    def on_train_end(self, args, state, control, **kwargs):
        if self.writer:
            self.writer.close()
            self.writer = None


class JsonlLoggerCallback(TrainerCallback):
    """Append every log dict to <output_dir>/train_log.jsonl (simple,
    dependency-free observability)."""

    def on_log(self, args, state, control, logs=None, **kwargs):
        if not state.is_world_process_zero or not logs:
            return
        os.makedirs(args.output_dir, exist_ok=True)
        path = os.path.join(args.output_dir, "train_log.jsonl")
        with open(path, "a") as f:
            f.write(json.dumps({"step": state.global_step, **logs}, default=str) + "\n")


INTEGRATION_TO_CALLBACK = {
    "tensorboard": TensorBoardCallback,
    "jsonl": JsonlLoggerCallback,
}


def get_reporting_integration_callbacks(report_to):
    callbacks = []
    for name in report_to or []:
        if name in INTEGRATION_TO_CALLBACK:
            callbacks.append(INTEGRATION_TO_CALLBACK[name]())
        else:
            logger.warning(f"Unknown report_to integration: {name}")
    return callbacks
