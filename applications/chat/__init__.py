from .dpo import DPOTrainer, dpo_loss, sequence_log_probs
from .ppo import GRPOTrainer, PPOTrainer, ValueCritic
from .reward_model import RewardModel, reward_pairwise_loss
from .sft import SFTTrainer

__all__ = ["SFTTrainer", "RewardModel", "reward_pairwise_loss", "DPOTrainer", "dpo_loss",
           "sequence_log_probs", "PPOTrainer", "ValueCritic", "GRPOTrainer"]
