"""Reward-model training entry point (reference: llm/alignment/rm)."""
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
from paddlenlp_amd.trl import RewardModel, RewardTrainer
from paddlenlp_amd.utils.log import logger


@dataclass
class ModelArgument:
    model_name_or_path: str = field(default=None)


@dataclass
class DataArgument:
    dataset_name_or_path: str = field(default=None)
    max_length: int = field(default=2048)


def convert_rm_example(ex, tokenizer, max_length):
    src = str(ex.get("src") or ex.get("prompt"))
    chosen = str(ex.get("chosen") or ex.get("tgt_chosen"))
    rejected = str(ex.get("rejected") or ex.get("tgt_rejected"))
    p = tokenizer.encode(src)
    out = {}
    for key, resp in (("chosen", chosen), ("rejected", rejected)):
        ids = (p + tokenizer.encode(resp))[:max_length]
        out[f"{key}_input_ids"] = ids
    return out


def _pad_batch(features):
    import torch as T

    out = {}
    for key in ("chosen_input_ids", "rejected_input_ids"):
        rows = [f[key] for f in features]
        maxlen = max(len(r) for r in rows)
        out[key] = T.tensor([r + [0] * (maxlen - len(r)) for r in rows],
                            dtype=T.long)
        out[key.replace("input_ids", "lens")] = T.tensor(
            [len(r) for r in rows], dtype=T.long)
    return out


def main():
    parser = PdArgumentParser((ModelArgument, DataArgument, TrainingArguments))
    model_args, data_args, training_args = parser.parse_json_file_and_cmd_lines()
    tokenizer = AutoTokenizer.from_pretrained(model_args.model_name_or_path)
    backbone = AutoModelForCausalLM.from_pretrained(model_args.model_name_or_path)
    if training_args.bf16:
        backbone = backbone.to(torch.bfloat16)
    model = RewardModel(backbone, backbone.config.hidden_size)

    train_ds = load_dataset(data_args.dataset_name_or_path, splits="train")
    train_ds = train_ds.map(partial(convert_rm_example, tokenizer=tokenizer,
                                    max_length=data_args.max_length))

    trainer = RewardTrainer(model=model, args=training_args,
                            train_dataset=train_ds, data_collator=_pad_batch,
                            tokenizer=tokenizer)
    trainer.train()
    if training_args.process_index == 0:
        os.makedirs(training_args.output_dir, exist_ok=True)
        torch.save(model.state_dict(),
                   os.path.join(training_args.output_dir, "reward_model.pt"))
        logger.info(f"reward model saved to {training_args.output_dir}")


if __name__ == "__main__":
    main()
