"""DPO entry point (reference: llm/alignment/dpo/run_dpo.py:292)."""
from __future__ import annotations

import os
import sys
from dataclasses import dataclass, field
from functools import partial

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))))

import torch

from paddlenlp_amd.datasets import load_dataset
from paddlenlp_amd.trainer import PdArgumentParser, TrainingArguments
from paddlenlp_amd.transformers import AutoModelForCausalLM, AutoTokenizer
from paddlenlp_amd.trl import DPOTrainer


@dataclass
class DPOArguments(TrainingArguments):
    beta: float = field(default=0.1)
    loss_type: str = field(default="sigmoid")
    label_smoothing: float = field(default=0.0)
    # reference-preset compat
    benchmark: bool = field(default=False)
    autotuner_benchmark: bool = field(default=False)
    ref_model_update_steps: int = field(default=-1)
    dpop_lambda: float = field(default=50.0)
    continue_training: bool = field(default=False)


@dataclass
class ModelArgument:
    model_name_or_path: str = field(default=None)
    # reference-preset compat (recompute granularity / flash attention
    # arrive via the shared trainer args)
    flash_mask: bool = field(default=False)
    # LoRA-DPO (reference dpo_lora presets): wraps the policy in a
    # LoRAModel before training
    lora: bool = field(default=False)
    lora_rank: int = field(default=8)
    lora_alpha: int = field(default=16)
    lora_dropout: float = field(default=0.0)
    rslora: bool = field(default=False)
    rslora_plus: bool = field(default=False)
    pissa: bool = field(default=False)


@dataclass
class DataArgument:
    dataset_name_or_path: str = field(default=None)
    max_length: int = field(default=2048)
    # reference-preset names
    train_dataset_path: str = field(default=None)
    dev_dataset_path: str = field(default=None)
    max_seq_len: int = field(default=0)
    max_prompt_len: int = field(default=0)
    lazy: bool = field(default=False)

    def __post_init__(self):
        if self.train_dataset_path and not self.dataset_name_or_path:
            self.dataset_name_or_path = self.train_dataset_path
        if self.max_seq_len and self.max_length == 2048:
            self.max_length = self.max_seq_len


def convert_dpo_example(ex, tokenizer, max_length):
    src = str(ex.get("src") or ex.get("prompt"))
    chosen = str(ex.get("chosen") or ex.get("tgt_chosen"))
    rejected = str(ex.get("rejected") or ex.get("tgt_rejected"))
    p = tokenizer.encode(src)
    out = {}
    for key, resp in (("chosen", chosen), ("rejected", rejected)):
        ids = (p + tokenizer.encode(resp))[:max_length]
        if tokenizer.eos_token_id is not None:
            ids = ids + [tokenizer.eos_token_id]
        labels = [-100] * (min(len(p), len(ids)) - 1) + ids[min(len(p), len(ids)):]
        out[f"{key}_input_ids"] = ids[:-1]
        out[f"{key}_labels"] = labels[:len(ids) - 1]
    return out


def main():
    parser = PdArgumentParser((ModelArgument, DataArgument, DPOArguments))
    model_args, data_args, training_args = parser.parse_json_file_and_cmd_lines()
    tokenizer = AutoTokenizer.from_pretrained(model_args.model_name_or_path)
    model = AutoModelForCausalLM.from_pretrained(model_args.model_name_or_path)
    if model_args.lora:
        from paddlenlp_amd.peft import LoRAConfig, LoRAModel

        model = LoRAModel(model, LoRAConfig(
            r=model_args.lora_rank, lora_alpha=model_args.lora_alpha,
            lora_dropout=model_args.lora_dropout, rslora=model_args.rslora))
        model.mark_only_lora_as_trainable()
    if training_args.bf16:
        model = model.to(torch.bfloat16)
    train_ds = load_dataset(data_args.dataset_name_or_path, splits="train")
    train_ds = train_ds.map(partial(convert_dpo_example, tokenizer=tokenizer,
                                    max_length=data_args.max_length))
    trainer = DPOTrainer(
        model=model, args=training_args, train_dataset=train_ds,
        tokenizer=tokenizer, beta=training_args.beta,
        loss_type=training_args.loss_type,
        label_smoothing=training_args.label_smoothing,
    )
    if training_args.do_train:
        trainer.train(resume_from_checkpoint=training_args.resume_from_checkpoint)
        trainer.save_model()


if __name__ == "__main__":
    main()
