"""PPO RLHF entry point (reference: llm/alignment/ppo/run_ppo.py).

Four-model loop: actor + frozen reference (deep-copied), critic value head,
and a reward model checkpoint (reward_model.pt from run_rm.py) or a
length-based stub reward for smoke runs.
"""
from __future__ import annotations

import copy
import os
import sys
from dataclasses import dataclass, field

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))))

import torch

from paddlenlp_amd.datasets import load_dataset
from paddlenlp_amd.trainer import PdArgumentParser, TrainingArguments
from paddlenlp_amd.transformers import AutoModelForCausalLM, AutoTokenizer
from paddlenlp_amd.trl import RewardModel
from paddlenlp_amd.trl.ppo_trainer import PPOConfig, PPOTrainer, ValueHeadModel
from paddlenlp_amd.utils.log import logger


@dataclass
class PPOArguments(TrainingArguments):
    kl_coef: float = field(default=0.1)
    clip_ratio: float = field(default=0.2)
    ppo_epochs: int = field(default=2)
    max_new_tokens: int = field(default=32)
    num_ppo_steps: int = field(default=10)
    rollout_batch_size: int = field(default=8)


@dataclass
class ModelArgument:
    model_name_or_path: str = field(default=None)
    reward_model_path: str = field(default=None)


@dataclass
class DataArgument:
    dataset_name_or_path: str = field(default=None)
    max_prompt_length: int = field(default=256)


def main():
    parser = PdArgumentParser((ModelArgument, DataArgument, PPOArguments))
    model_args, data_args, training_args = parser.parse_json_file_and_cmd_lines()
    tokenizer = AutoTokenizer.from_pretrained(model_args.model_name_or_path)
    actor = AutoModelForCausalLM.from_pretrained(model_args.model_name_or_path)
    reference = copy.deepcopy(actor)
    critic = ValueHeadModel(copy.deepcopy(actor), actor.config.hidden_size)

    if model_args.reward_model_path:
        rm_backbone = AutoModelForCausalLM.from_pretrained(
            model_args.model_name_or_path)
        reward_model = RewardModel(rm_backbone, rm_backbone.config.hidden_size)
        reward_model.load_state_dict(torch.load(
            os.path.join(model_args.reward_model_path, "reward_model.pt"),
            weights_only=True))
        reward_model.eval()

        def reward_fn(prompt_ids, response_ids):
            ids = torch.cat([prompt_ids, response_ids])[None]
            return float(reward_model.score(ids))
    else:
        logger.warning("no reward_model_path: using a length-stub reward")

        def reward_fn(prompt_ids, response_ids):
            return float(min(len(response_ids), 8)) / 8.0

    cfg = PPOConfig(
        kl_coef=training_args.kl_coef, clip_ratio=training_args.clip_ratio,
        ppo_epochs=training_args.ppo_epochs,
        max_new_tokens=training_args.max_new_tokens,
        learning_rate=training_args.learning_rate)
    ppo = PPOTrainer(actor, critic, reference, reward_fn,
                     tokenizer=tokenizer, config=cfg)

    train_ds = load_dataset(data_args.dataset_name_or_path, splits="train")
    prompts = [str(ex.get("src") or ex.get("prompt")) for ex in train_ds]
    B = training_args.rollout_batch_size
    for step in range(training_args.num_ppo_steps):
        batch = [prompts[(step * B + i) % len(prompts)] for i in range(B)]
        enc = tokenizer(batch, padding=True, return_tensors="pt")
        ids = enc["input_ids"][:, -data_args.max_prompt_length:]
        stats = ppo.step(ids)
        logger.info(f"ppo step {step}: {stats}")

    if training_args.process_index == 0:
        os.makedirs(training_args.output_dir, exist_ok=True)
        actor.save_pretrained(training_args.output_dir)
        logger.info(f"PPO actor saved to {training_args.output_dir}")


if __name__ == "__main__":
    main()
