from .dpo_criterion import DPOCriterion  # noqa: F401
from .dpo_trainer import DPOTrainer, sequence_logprob  # noqa: F401
from .ppo_trainer import PPOConfig, PPOTrainer, ValueHeadModel  # noqa: F401
from .reward_trainer import RewardModel, RewardTrainer  # noqa: F401
