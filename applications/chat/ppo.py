"""PPO trainer for RLHF (reference:
applications/ColossalChat/coati/trainer/ppo.py + experience_maker).

Lean but complete actor-critic loop: rollouts sampled from the actor,
per-token log-probs against a frozen reference for the KL shaping, a
value head for the baseline, GAE advantages, clipped surrogate + clipped
value loss. The reward can be a trained ``RewardModel`` or any callable
``f(sequences, prompt_len) -> [B]`` (tests use a synthetic preference).
"""

import copy
from typing import Callable, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from colossalai_amd import Booster

from .dpo import sequence_log_probs  # noqa: F401  (re-exported convenience)

__all__ = ["PPOTrainer", "ValueCritic", "GRPOTrainer"]


class ValueCritic(nn.Module):
    """Per-token value head over a causal-LM backbone."""

    def __init__(self, backbone: nn.Module):
        super().__init__()
        self.model = copy.deepcopy(backbone.model)
        h = backbone.config.hidden_size
        self.value_head = nn.Linear(h, 1, bias=False)
        self.value_head.weight.data.normal_(0.0, 1.0 / (h + 1) ** 0.5)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        return self.value_head(self.model(input_ids)).squeeze(-1)  # [B, S]


def _token_logprobs(logits: torch.Tensor, ids: torch.Tensor) -> torch.Tensor:
    """log p(ids[t] | ids[<t]) for t >= 1 -> [B, S-1]."""
    lp = F.log_softmax(logits[:, :-1].float(), dim=-1)
    return lp.gather(-1, ids[:, 1:].unsqueeze(-1)).squeeze(-1)


class PPOTrainer:
    def __init__(
        self,
        actor,
        critic: ValueCritic,
        reward_fn: Callable,
        actor_optimizer,
        critic_optimizer,
        booster: Optional[Booster] = None,
        clip_eps: float = 0.2,
        value_clip: float = 0.2,
        kl_coef: float = 0.02,
        gamma: float = 1.0,
        lam: float = 0.95,
        ppo_epochs: int = 2,
    ):
        booster = booster or Booster()
        criterion = lambda out, batch: out
        self.actor, self.actor_opt, *_ = booster.boost(actor, actor_optimizer, criterion)
        self.critic, self.critic_opt, *_ = booster.boost(critic, critic_optimizer, criterion)
        self.booster = booster
        self.ref = copy.deepcopy(actor).eval()
        for p in self.ref.parameters():
            p.requires_grad_(False)
        self.reward_fn = reward_fn
        self.clip_eps = clip_eps
        self.value_clip = value_clip
        self.kl_coef = kl_coef
        self.gamma = gamma
        self.lam = lam
        self.ppo_epochs = ppo_epochs

    # ------------------------------------------------------------- rollouts
    @torch.no_grad()
    def _rollout(self, prompts: torch.Tensor, max_new_tokens: int, temperature: float = 1.0):
        """Sample continuations; -> sequences [B, P+N]."""
        seq = prompts
        for _ in range(max_new_tokens):
            logits = self.actor(input_ids=seq)["logits"][:, -1]
            probs = torch.softmax(logits.float() / max(temperature, 1e-5), dim=-1)
            seq = torch.cat([seq, torch.multinomial(probs, 1)], dim=1)
        return seq

    @torch.no_grad()
    def make_experience(self, prompts: torch.Tensor, max_new_tokens: int):
        P = prompts.shape[1]
        seq = self._rollout(prompts, max_new_tokens)
        old_logp = _token_logprobs(self.actor(input_ids=seq)["logits"], seq)[:, P - 1 :]
        ref_logp = _token_logprobs(self.ref(input_ids=seq)["logits"], seq)[:, P - 1 :]
        values = self.critic(seq)[:, P - 1 : -1]  # value before each action
        reward = self.reward_fn(seq, P).float()  # [B] terminal reward

        # per-token reward: -kl_coef * KL estimate, terminal reward on last
        rewards = -self.kl_coef * (old_logp - ref_logp)
        rewards[:, -1] += reward

        # GAE over the response tokens
        B, N = rewards.shape
        adv = torch.zeros_like(rewards)
        last = torch.zeros(B)
        next_v = torch.zeros(B)
        for t in range(N - 1, -1, -1):
            delta = rewards[:, t] + self.gamma * next_v - values[:, t]
            last = delta + self.gamma * self.lam * last
            adv[:, t] = last
            next_v = values[:, t]
        returns = adv + values
        adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        return {"seq": seq, "old_logp": old_logp, "values": values,
                "adv": adv, "returns": returns, "prompt_len": P,
                "mean_reward": float(reward.mean())}

    # --------------------------------------------------------------- update
    def train_step(self, prompts: torch.Tensor, max_new_tokens: int = 8):
        exp = self.make_experience(prompts, max_new_tokens)
        P, seq = exp["prompt_len"], exp["seq"]
        stats = {}
        for _ in range(self.ppo_epochs):
            logp = _token_logprobs(self.actor(input_ids=seq)["logits"], seq)[:, P - 1 :]
            ratio = (logp - exp["old_logp"]).exp()
            s1 = ratio * exp["adv"]
            s2 = ratio.clamp(1 - self.clip_eps, 1 + self.clip_eps) * exp["adv"]
            policy_loss = -torch.min(s1, s2).mean()
            self.actor_opt.zero_grad()
            self.booster.backward(policy_loss, self.actor_opt)
            self.actor_opt.step()

            v = self.critic(seq)[:, P - 1 : -1]
            v_clip = exp["values"] + (v - exp["values"]).clamp(-self.value_clip, self.value_clip)
            value_loss = torch.max((v - exp["returns"]) ** 2, (v_clip - exp["returns"]) ** 2).mean()
            self.critic_opt.zero_grad()
            self.booster.backward(value_loss, self.critic_opt)
            self.critic_opt.step()
            stats = {"policy_loss": float(policy_loss), "value_loss": float(value_loss),
                     "reward": exp["mean_reward"]}
        return stats


class GRPOTrainer(PPOTrainer):
    """Group Relative Policy Optimization (reference:
    applications/ColossalChat GRPO trainer; Shao et al., DeepSeekMath).

    Critic-free: for each prompt, sample ``group_size`` continuations and
    use the group-normalized reward as a per-sequence advantage (broadcast
    over its tokens). Keeps PPO's clipped surrogate and the reference-KL
    shaping; drops the value network entirely."""

    def __init__(self, actor, reward_fn, actor_optimizer, booster=None,
                 group_size: int = 4, clip_eps: float = 0.2, kl_coef: float = 0.02,
                 ppo_epochs: int = 2):
        booster = booster or Booster()
        criterion = lambda out, batch: out
        self.actor, self.actor_opt, *_ = booster.boost(actor, actor_optimizer, criterion)
        self.booster = booster
        self.ref = copy.deepcopy(actor).eval()
        for p in self.ref.parameters():
            p.requires_grad_(False)
        self.reward_fn = reward_fn
        self.group_size = group_size
        self.clip_eps = clip_eps
        self.kl_coef = kl_coef
        self.ppo_epochs = ppo_epochs

    def train_step(self, prompts: torch.Tensor, max_new_tokens: int = 8):
        B, P = prompts.shape
        G = self.group_size
        rep = prompts.repeat_interleave(G, dim=0)  # [B*G, P]
        with torch.no_grad():
            seq = self._rollout(rep, max_new_tokens)
            old_logp = _token_logprobs(self.actor(input_ids=seq)["logits"], seq)[:, P - 1 :]
            ref_logp = _token_logprobs(self.ref(input_ids=seq)["logits"], seq)[:, P - 1 :]
            reward = self.reward_fn(seq, P).float().view(B, G)
            adv = (reward - reward.mean(dim=1, keepdim=True)) / (reward.std(dim=1, keepdim=True) + 1e-6)
            adv = adv.view(B * G, 1)  # broadcast over the sequence's tokens

        stats = {}
        for _ in range(self.ppo_epochs):
            logp = _token_logprobs(self.actor(input_ids=seq)["logits"], seq)[:, P - 1 :]
            ratio = (logp - old_logp).exp()
            s1 = ratio * adv
            s2 = ratio.clamp(1 - self.clip_eps, 1 + self.clip_eps) * adv
            kl = (old_logp - logp)  # k1 estimator vs the sampling policy
            ref_kl = (logp - ref_logp)
            loss = (-torch.min(s1, s2) + self.kl_coef * ref_kl).mean()
            self.actor_opt.zero_grad()
            self.booster.backward(loss, self.actor_opt)
            self.actor_opt.step()
            stats = {"policy_loss": float(loss), "reward": float(reward.mean()),
                     "kl": float(kl.mean())}
        return stats
