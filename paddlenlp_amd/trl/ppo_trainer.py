"""PPO trainer: actor/critic/reference/reward 4-model RLHF loop.

Reference behavior: llm/alignment/ppo/ppo_trainer.py (1.8k LoC) — rollout via
generation, per-token KL penalty against the frozen reference, GAE
advantages, clipped surrogate policy loss + clipped value loss over
minibatch epochs.  This is the compact MI355X-native form: generation runs
through GenerationMixin (or the fused inference engine when provided), and
the update steps use the framework's FusedAdamW.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Optional

import torch

from ..generation import GenerationConfig
from ..trainer.optimizer import FusedAdamW


@dataclass
class PPOConfig:
    learning_rate: float = 1e-6
    critic_learning_rate: float = 1e-5
    kl_coef: float = 0.1
    gamma: float = 1.0
    gae_lambda: float = 0.95
    clip_ratio: float = 0.2
    value_clip: float = 0.2
    ppo_epochs: int = 2
    minibatch_size: int = 4
    max_new_tokens: int = 32
    temperature: float = 1.0
    top_p: float = 1.0
    vf_coef: float = 0.5


class ValueHeadModel(torch.nn.Module):
    """Backbone + scalar value head (the critic)."""

    def __init__(self, backbone, hidden_size: int):
        super().__init__()
        self.backbone = backbone
        self.value_head = torch.nn.Linear(hidden_size, 1, bias=False)

    def forward(self, input_ids):
        base = getattr(self.backbone, self.backbone.base_model_prefix, self.backbone)
        hidden = base(input_ids=input_ids)
        if isinstance(hidden, tuple):
            hidden = hidden[0]
        return self.value_head(hidden).squeeze(-1)  # [B, S]


class PPOTrainer:
    def __init__(self, actor, critic: ValueHeadModel, reference, reward_fn: Callable,
                 tokenizer=None, config: Optional[PPOConfig] = None):
        """reward_fn(query_ids, response_ids) -> scalar reward per sequence."""
        self.cfg = config or PPOConfig()
        self.actor = actor
        self.critic = critic
        self.reference = reference
        for p in self.reference.parameters():
            p.requires_grad_(False)
        self.reference.eval()
        self.reward_fn = reward_fn
        self.tokenizer = tokenizer
        self.actor_opt = FusedAdamW(
            [p for p in actor.parameters() if p.requires_grad],
            lr=self.cfg.learning_rate, master_weights=False)
        self.critic_opt = FusedAdamW(
            [p for p in critic.parameters() if p.requires_grad],
            lr=self.cfg.critic_learning_rate, master_weights=False)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def rollout(self, prompt_ids: torch.Tensor):
        """Generate responses; collect old logprobs, values and rewards."""
        cfg = self.cfg
        self.actor.eval()
        gen = GenerationConfig(
            max_new_tokens=cfg.max_new_tokens, do_sample=True,
            temperature=cfg.temperature, top_p=cfg.top_p,
            eos_token_id=getattr(self.tokenizer, "eos_token_id", None) if self.tokenizer else None,
            pad_token_id=0,
        )
        responses, _ = self.actor.generate(prompt_ids, gen)
        full = torch.cat([prompt_ids, responses], dim=1)
        B, P = prompt_ids.shape
        R = responses.shape[1]

        # token-level logprobs of the response region
        labels = torch.full_like(full, -100)
        labels[:, P:] = full[:, P:]
        shifted_labels = labels[:, 1:]

        def token_logps(model):
            logits = model(input_ids=full)
            if isinstance(logits, tuple):
                logits = logits[0]
            lp = logits[:, :-1].float().log_softmax(-1)
            safe = shifted_labels.clamp(min=0)
            tok = lp.gather(-1, safe[..., None]).squeeze(-1)
            return tok * (shifted_labels != -100)

        old_logps = token_logps(self.actor)
        ref_logps = token_logps(self.reference)
        values = self.critic(full)[:, :-1] * (shifted_labels != -100)

        # rewards: per-token KL penalty + terminal reward from the reward fn
        kl = old_logps - ref_logps
        rewards = -self.cfg.kl_coef * kl
        final = torch.tensor(
            [float(self.reward_fn(prompt_ids[b], responses[b])) for b in range(B)],
            device=full.device)
        mask = (shifted_labels != -100).float()
        last_idx = (mask.cumsum(-1) * mask).argmax(-1)  # last response position
        rewards[torch.arange(B, device=full.device), last_idx] += final

        return {
            "full": full, "labels": shifted_labels, "mask": mask,
            "old_logps": old_logps, "values": values, "rewards": rewards,
            "kl": (kl * mask).sum() / mask.sum().clamp(min=1),
            "reward_mean": final.mean(),
        }

    @staticmethod
    def compute_gae(rewards, values, mask, gamma, lam):
        B, S = rewards.shape
        adv = torch.zeros_like(rewards)
        last = torch.zeros(B, device=rewards.device)
        for t in reversed(range(S)):
            next_v = values[:, t + 1] if t + 1 < S else torch.zeros_like(last)
            delta = rewards[:, t] + gamma * next_v - values[:, t]
            last = delta + gamma * lam * last
            adv[:, t] = last
        adv = adv * mask
        returns = adv + values
        # advantage whitening over response tokens
        n = mask.sum().clamp(min=1)
        mean = (adv * mask).sum() / n
        var = ((adv - mean) ** 2 * mask).sum() / n
        adv = (adv - mean) / (var.sqrt() + 1e-8) * mask
        return adv, returns

    # ------------------------------------------------------------------
    def step(self, prompt_ids: torch.Tensor) -> dict:
        cfg = self.cfg
        data = self.rollout(prompt_ids)
        adv, returns = self.compute_gae(
            data["rewards"], data["values"], data["mask"], cfg.gamma, cfg.gae_lambda)

        full, labels, mask = data["full"], data["labels"], data["mask"]
        old_logps, old_values = data["old_logps"], data["values"]
        B = full.shape[0]
        stats = {"policy_loss": 0.0, "value_loss": 0.0, "kl": float(data["kl"]),
                 "reward_mean": float(data["reward_mean"])}
        n_updates = 0
        self.actor.train()
        self.critic.train()
        for _ in range(cfg.ppo_epochs):
            perm = torch.randperm(B)
            for s in range(0, B, cfg.minibatch_size):
                idx = perm[s:s + cfg.minibatch_size]
                mb_full = full[idx]
                mb_labels = labels[idx]
                mb_mask = mask[idx]

                logits = self.actor(input_ids=mb_full)
                if isinstance(logits, tuple):
                    logits = logits[0]
                lp = logits[:, :-1].float().log_softmax(-1)
                tok = lp.gather(-1, mb_labels.clamp(min=0)[..., None]).squeeze(-1) * mb_mask

                ratio = torch.exp(tok - old_logps[idx])
                surr1 = ratio * adv[idx]
                surr2 = torch.clamp(ratio, 1 - cfg.clip_ratio, 1 + cfg.clip_ratio) * adv[idx]
                n = mb_mask.sum().clamp(min=1)
                policy_loss = -(torch.min(surr1, surr2) * mb_mask).sum() / n

                self.actor_opt.zero_grad(set_to_none=True)
                policy_loss.backward()
                self.actor_opt.step()

                v = self.critic(mb_full)[:, :-1]
                v_clip = old_values[idx] + torch.clamp(
                    v - old_values[idx], -cfg.value_clip, cfg.value_clip)
                vloss = torch.max((v - returns[idx]) ** 2,
                                  (v_clip - returns[idx]) ** 2)
                value_loss = cfg.vf_coef * (vloss * mb_mask).sum() / n

                self.critic_opt.zero_grad(set_to_none=True)
                value_loss.backward()
                self.critic_opt.step()

                stats["policy_loss"] += float(policy_loss)
                stats["value_loss"] += float(value_loss)
                n_updates += 1
        stats["policy_loss"] /= max(1, n_updates)
        stats["value_loss"] /= max(1, n_updates)
        return stats
