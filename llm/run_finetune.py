"""SFT / LoRA / prefix fine-tuning entry point.

Reference behavior: llm/run_finetune.py main :77 — same trainer skeleton as
pretrain with: load_dataset + chat-template tokenization (llm/utils/data.py),
ZeroPadding packing (:385-412), FlashMask via attn_mask_startend_row_indices,
PEFT wrapping (LoRA/prefix) before the Trainer.
"""
from __future__ import annotations

import os
import sys
from dataclasses import dataclass, field
from functools import partial
from typing import Optional

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from paddlenlp_amd.data import DataCollatorWithPadding
from paddlenlp_amd.datasets import ZeroPaddingMapDataset, load_dataset
from paddlenlp_amd.peft import LoRAConfig, LoRAModel, PrefixConfig, PrefixModelForCausalLM
from paddlenlp_amd.trainer import PdArgumentParser, Trainer, TrainingArguments
from paddlenlp_amd.transformers import (
    AutoConfig,
    AutoModelForCausalLM,
    AutoTokenizer,
    LlmMetaConfig,
)
from paddlenlp_amd.transformers.model_utils import unwrap_model
from paddlenlp_amd.utils.log import logger
from utils.data import convert_example


@dataclass
class FinetuneArguments(TrainingArguments):
    # reference-preset compat: generation-based eval (BLEU/Rouge over
    # generate()) — loss/accuracy eval runs when False (the default in
    # most presets); True falls back to loss-eval with a warning
    eval_with_do_generation: bool = False
    # benchmark-mode flags (reference uses them to strip callbacks for
    # throughput runs; accepted, benchmarking here is tools/bench_sft.py)
    benchmark: bool = False
    autotuner_benchmark: bool = False
    # PTQ flow flags (reference QuantArgument; wired to
    # trainer_compress.post_training_quantization)
    do_ptq: bool = False
    ptq_step: int = 32
    do_gptq: bool = False
    gptq_step: int = 8
    smooth: bool = False
    smooth_step: int = 16
    smooth_all_linears: bool = False
    smooth_piecewise_search: bool = False
    smooth_k_piece: int = 3
    smooth_search_piece: bool = False
    auto_clip: bool = False
    autoclip_step: int = 8
    quant_type: str = "a8w8"
    weight_quant_method: str = "abs_max_channel_wise"
    act_quant_method: str = "avg"
    cachekv_quant_method: str = "abs_max_headwise"
    do_awq: bool = False
    # resume base-model training from an SFT checkpoint (reference flag)
    continue_training: bool = False
    pass


@dataclass
class ModelArgument:
    model_name_or_path: str = field(default=None)
    lora: bool = field(default=False)
    lora_rank: int = field(default=8)
    # reference-preset compat (rsLoRA+ scales the B learning rate)
    rslora_plus: bool = field(default=False)
    pissa: bool = field(default=False)
    vera: bool = field(default=False)
    vera_rank: int = field(default=8)
    lora_alpha: float = field(default=16.0)
    lora_dropout: float = field(default=0.0)
    rslora: bool = field(default=False)
    prefix_tuning: bool = field(default=False)
    num_prefix_tokens: int = field(default=16)
    flash_mask: bool = field(default=False)
    # QLoRA: quantize the frozen base to 4-bit before wrapping with LoRA
    # (reference run_finetune weight_quantize_algo; "" disables)
    weight_quantize_algo: str = field(default="")


@dataclass
class DataArgument:
    dataset_name_or_path: str = field(default=None)
    max_length: int = field(default=2048)
    # max prompt (source) length; the target gets the remaining budget
    # (reference run_finetune DataArgument src_length)
    src_length: int = field(default=1024)
    zero_padding: bool = field(default=False)
    # reference-preset compat: lazy dataset loading is the default here
    # (map-style loading is strict), the flag is accepted for parity
    lazy: bool = field(default=False)
    pad_to_multiple_of: int = field(default=0)


def main():
    parser = PdArgumentParser((ModelArgument, DataArgument, FinetuneArguments))
    model_args, data_args, training_args = parser.parse_json_file_and_cmd_lines()

    config = AutoConfig.from_pretrained(model_args.model_name_or_path)
    LlmMetaConfig.set_llm_config(config, training_args)
    config.tensor_parallel_rank = training_args.topology.get_rank_in("mp")
    config.dtype = "bfloat16" if training_args.bf16 else "float32"

    tokenizer = AutoTokenizer.from_pretrained(model_args.model_name_or_path)
    model = AutoModelForCausalLM.from_pretrained(model_args.model_name_or_path, config=config)
    if training_args.bf16:
        model = model.to(torch.bfloat16)

    if model_args.flash_mask and not data_args.zero_padding:
        raise ValueError("flash_mask requires zero_padding (reference run_finetune.py:182-184)")

    if model_args.weight_quantize_algo:
        from paddlenlp_amd.quantization import (
            QuantizationConfig, replace_with_quantization_linear)

        replace_with_quantization_linear(
            model, QuantizationConfig(
                weight_quantize_algo=model_args.weight_quantize_algo))
        logger.info(f"base model quantized with {model_args.weight_quantize_algo}")

    if model_args.lora:
        lora_config = LoRAConfig(
            r=model_args.lora_rank, lora_alpha=model_args.lora_alpha,
            lora_dropout=model_args.lora_dropout, rslora=model_args.rslora,
        )
        model = LoRAModel(model, lora_config)
    elif model_args.prefix_tuning:
        model = PrefixModelForCausalLM(
            model, PrefixConfig(num_prefix_tokens=model_args.num_prefix_tokens))

    train_ds = load_dataset(data_args.dataset_name_or_path, splits="train")
    dev_ds = None
    try:
        dev_ds = load_dataset(data_args.dataset_name_or_path, splits="dev")
    except (FileNotFoundError, TypeError):
        pass

    trans_fn = partial(convert_example, tokenizer=tokenizer,
                       max_length=data_args.max_length,
                       src_length=data_args.src_length)
    train_ds = train_ds.map(trans_fn)
    if dev_ds is not None:
        dev_ds = dev_ds.map(trans_fn)

    if data_args.zero_padding:
        train_ds = ZeroPaddingMapDataset(train_ds, tokenizer=tokenizer,
                                         max_length=data_args.max_length)
        if dev_ds is not None:
            dev_ds = ZeroPaddingMapDataset(dev_ds, tokenizer=tokenizer,
                                           max_length=data_args.max_length)
        collator = None  # already fixed-shape
    else:
        collator = DataCollatorWithPadding(tokenizer=tokenizer, max_length=data_args.max_length)

    trainer = Trainer(
        model=model,
        args=training_args,
        train_dataset=train_ds,
        eval_dataset=dev_ds,
        data_collator=collator,
        tokenizer=tokenizer,
    )
    if training_args.do_train:
        trainer.train(resume_from_checkpoint=training_args.resume_from_checkpoint)
        if model_args.lora or model_args.prefix_tuning:
            if training_args.process_index == 0:
                model.save_pretrained(training_args.output_dir)
                tokenizer.save_pretrained(training_args.output_dir)
        else:
            trainer.save_model()
    if training_args.do_eval and dev_ds is not None:
        trainer.evaluate()

    if training_args.do_ptq:
        # PTQ flow (reference run_finetune.py:629-670 apply_ptq): calibrate
        # activation scales over up to ptq_step calibration batches and
        # replace linears with simulated a8w8 modules, then save
        from paddlenlp_amd.trainer.trainer_compress import (
            post_training_quantization,
        )

        calib_loader = trainer.get_eval_dataloader(dev_ds or train_ds)
        post_training_quantization(
            unwrap_model(trainer.model), calib_loader,
            algo=training_args.act_quant_method,
            batch_nums=training_args.ptq_step)
        if training_args.process_index == 0:
            out = os.path.join(training_args.output_dir, "ptq")
            os.makedirs(out, exist_ok=True)
            torch.save(unwrap_model(trainer.model).state_dict(),
                       os.path.join(out, "quantized_model.pt"))
            logger.info(f"PTQ model saved to {out}")


if __name__ == "__main__":
    main()
