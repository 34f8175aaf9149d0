"""Causal-LM pretraining entry point.

Reference behavior: llm/run_pretrain.py (main :358 — parse JSON+CLI args,
build tokenizer/config, LlmMetaConfig.set_llm_config :415, model_class
from_config :479-501 with the Pipe variant when pp>1 :480, pretraining
dataset :539, PretrainingTrainer :555, trainer.train :573).

Usage:
  python -m torch.distributed.run --nproc-per-node N llm/run_pretrain.py config.json
  python llm/run_pretrain.py config.json --max_steps 100
"""
from __future__ import annotations

import math
import os
import sys
from dataclasses import dataclass, field
from typing import Optional

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from paddlenlp_amd.data import DataCollatorForLanguageModeling
from paddlenlp_amd.data.causal_dataset import build_train_valid_test_datasets
from paddlenlp_amd.trainer import (
    PdArgumentParser,
    Trainer,
    TrainingArguments,
    get_last_checkpoint,
    speed_metrics,
)
from paddlenlp_amd.transformers import (
    AutoConfig,
    AutoModelForCausalLM,
    AutoTokenizer,
    LlmMetaConfig,
)
from paddlenlp_amd.utils.log import logger


@dataclass
class PreTrainingArguments(TrainingArguments):
    min_learning_rate: float = field(default=1e-5)
    # start from a released checkpoint instead of random init (reference)
    continue_training: bool = field(default=False)
    decay_steps: int = field(default=0)

    def __post_init__(self):
        super().__post_init__()
        if self.min_learning_rate and self.learning_rate:
            self.min_lr_ratio = self.min_learning_rate / self.learning_rate


@dataclass
class ModelArguments:
    model_name_or_path: str = field(default=None)
    config_name: Optional[str] = field(default=None)
    tokenizer_name_or_path: Optional[str] = field(default=None)
    num_hidden_layers: Optional[int] = field(default=None)
    hidden_dropout_prob: float = field(default=0.0)
    fuse_attention_qkv: bool = field(default=True)
    fuse_attention_ffn: bool = field(default=True)


@dataclass
class DataArguments:
    input_dir: str = field(default=None)
    split: str = field(default="949,50,1")
    max_seq_length: int = field(default=4096)
    data_cache: Optional[str] = field(default=None)


def create_pretrained_dataset(data_args, training_args, tokenizer=None):
    """Reference: run_pretrain.py create_pretrained_dataset :539."""
    # find .bin/.idx prefixes under input_dir
    prefixes = []
    for f in sorted(os.listdir(data_args.input_dir)):
        if f.endswith(".bin"):
            prefixes.append(os.path.join(data_args.input_dir, f[:-4]))
    if not prefixes:
        raise FileNotFoundError(f"No .bin/.idx data under {data_args.input_dir}")

    max_steps = max(training_args.max_steps, 1)
    train_samples = max_steps * training_args.global_train_batch_size
    eval_samples = max(64, training_args.per_device_eval_batch_size * 8)

    train_ds, valid_ds, test_ds = build_train_valid_test_datasets(
        prefixes if len(prefixes) > 1 else prefixes[0],
        data_args.split,
        [train_samples, eval_samples, eval_samples],
        data_args.max_seq_length,
        training_args.seed,
        data_cache_path=data_args.data_cache,
    )
    return train_ds, valid_ds, test_ds


def main():
    parser = PdArgumentParser((ModelArguments, DataArguments, PreTrainingArguments))
    model_args, data_args, training_args = parser.parse_json_file_and_cmd_lines()

    training_args.print_config()

    config = AutoConfig.from_pretrained(model_args.config_name or model_args.model_name_or_path)
    LlmMetaConfig.set_llm_config(config, training_args)
    config.tensor_parallel_rank = training_args.topology.get_rank_in("mp")
    if model_args.num_hidden_layers is not None:
        config.num_hidden_layers = model_args.num_hidden_layers
    config.fuse_attention_qkv = model_args.fuse_attention_qkv
    config.fuse_attention_ffn = model_args.fuse_attention_ffn
    config.dtype = "bfloat16" if training_args.bf16 else (
        "float16" if training_args.fp16 else "float32")

    tokenizer = None
    tok_path = model_args.tokenizer_name_or_path or model_args.model_name_or_path
    try:
        tokenizer = AutoTokenizer.from_pretrained(tok_path)
    except FileNotFoundError:
        logger.warning(f"No tokenizer found at {tok_path}; continuing without one")

    if training_args.pipeline_parallel_degree > 1:
        from paddlenlp_amd.transformers.llama.modeling_pp import (
            LlamaForCausalLMPipe,
            MistralForCausalLMPipe,
            Qwen2ForCausalLMPipe,
        )

        pipe_registry = {"llama": LlamaForCausalLMPipe,
                         "qwen2": Qwen2ForCausalLMPipe,
                         "mistral": MistralForCausalLMPipe}
        assert config.model_type in pipe_registry, \
            f"no pipeline variant for model_type {config.model_type!r}"
        model = pipe_registry[config.model_type](
            config, num_virtual_stages=training_args.virtual_pp_degree)
        if training_args.bf16:
            model = model.to(torch.bfloat16)
    else:
        model = AutoModelForCausalLM.from_config(config)

    train_ds, valid_ds, test_ds = create_pretrained_dataset(data_args, training_args, tokenizer)

    trainer = Trainer(
        model=model,
        args=training_args,
        train_dataset=train_ds,
        eval_dataset=valid_ds,
        data_collator=DataCollatorForLanguageModeling(tokenizer),
        tokenizer=tokenizer,
    )

    checkpoint = None
    if training_args.resume_from_checkpoint:
        checkpoint = training_args.resume_from_checkpoint
    elif not training_args.overwrite_output_dir:
        checkpoint = get_last_checkpoint(training_args.output_dir)

    if training_args.do_train:
        result = trainer.train(resume_from_checkpoint=checkpoint)
        metrics = result.metrics
        total_tokens = trainer.state.consumed_samples * data_args.max_seq_length
        if metrics.get("train_runtime"):
            metrics["effective_tokens_per_second"] = round(
                total_tokens / metrics["train_runtime"], 2)
            logger.info(f"ips: {metrics['effective_tokens_per_second']} tokens/s")
        trainer.save_model()
        if trainer.args.should_log:
            logger.info(f"train metrics: {metrics}")

    if training_args.do_eval and valid_ds is not None:
        trainer.evaluate()


if __name__ == "__main__":
    main()
