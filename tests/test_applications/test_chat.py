"""Chat-alignment application: SFT, reward model, DPO (CPU, plugin-free
Booster; the trainers accept any plugin)."""

import torch

from applications.chat import DPOTrainer, RewardModel, SFTTrainer, reward_pairwise_loss
from colossalai_amd import Booster
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM


def _tiny():
    return LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64, num_hidden_layers=2,
                       num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=32)


def test_sft_trainer_learns():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny())
    trainer = SFTTrainer(model, torch.optim.AdamW(model.parameters(), lr=5e-3), Booster())
    x = torch.randint(0, 64, (4, 16))
    labels = x.clone()
    labels[:, :8] = -100  # prompt masked
    batch = {"input_ids": x, "labels": labels}
    losses = [trainer.train_step(batch) for _ in range(15)]
    assert losses[-1] < losses[0] * 0.5, losses


def test_reward_model_ranks():
    torch.manual_seed(0)
    rm = RewardModel(LlamaForCausalLM(_tiny()))
    opt = torch.optim.AdamW(rm.parameters(), lr=5e-3)
    chosen = torch.randint(0, 32, (4, 12))
    rejected = torch.randint(32, 64, (4, 12))
    mask = torch.ones(4, 12, dtype=torch.long)
    mask[:, -3:] = 0
    for _ in range(30):
        loss = reward_pairwise_loss(rm(chosen, mask), rm(rejected, mask))
        opt.zero_grad(); loss.backward(); opt.step()
    with torch.no_grad():
        assert (rm(chosen, mask) > rm(rejected, mask)).all(), "reward model failed to rank pairs"


def test_dpo_trainer_prefers_chosen():
    torch.manual_seed(0)
    policy = LlamaForCausalLM(_tiny())
    trainer = DPOTrainer(policy, torch.optim.AdamW(policy.parameters(), lr=5e-3), Booster(), beta=0.5)
    chosen = torch.randint(0, 32, (4, 12))
    rejected = torch.randint(32, 64, (4, 12))
    mask = torch.ones(4, 12, dtype=torch.long)
    mask[:, :4] = 0  # shared "prompt"
    batch = {"chosen_ids": chosen, "chosen_mask": mask,
             "rejected_ids": rejected, "rejected_mask": mask}
    hist = trainer.fit([batch] * 20, epochs=1)
    losses = [h[0] for h in hist]
    accs = [h[1] for h in hist]
    assert losses[-1] < losses[0] * 0.5, losses
    assert accs[-1] == 1.0, accs


def test_ppo_trainer_shifts_policy_toward_reward():
    """Synthetic preference: reward = fraction of generated tokens < 32.
    After PPO steps, the actor must generate low tokens much more often."""
    from applications.chat import PPOTrainer, ValueCritic

    torch.manual_seed(0)
    actor = LlamaForCausalLM(_tiny())
    critic = ValueCritic(actor)

    def reward_fn(seq, prompt_len):
        gen = seq[:, prompt_len:]
        return (gen < 32).float().mean(dim=1) * 2 - 1  # in [-1, 1]

    trainer = PPOTrainer(actor, critic, reward_fn,
                         torch.optim.AdamW(actor.parameters(), lr=5e-3),
                         torch.optim.AdamW(critic.parameters(), lr=5e-3),
                         kl_coef=0.0, ppo_epochs=3)
    prompts = torch.randint(0, 64, (16, 6))

    def low_frac():
        with torch.no_grad():
            fr = [float((trainer._rollout(prompts, 8)[:, 6:] < 32).float().mean()) for _ in range(4)]
        return sum(fr) / len(fr)

    before = low_frac()
    rewards = []
    for _ in range(25):
        stats = trainer.train_step(prompts, max_new_tokens=8)
        rewards.append(stats["reward"])
    after = low_frac()
    assert after > max(before + 0.2, 0.8), (
        f"policy did not shift: {before:.2f} -> {after:.2f} (rewards {rewards})")


def test_grpo_trainer_shifts_policy():
    from applications.chat import GRPOTrainer

    torch.manual_seed(1)
    actor = LlamaForCausalLM(_tiny())

    def reward_fn(seq, prompt_len):
        gen = seq[:, prompt_len:]
        return (gen < 32).float().mean(dim=1) * 2 - 1

    trainer = GRPOTrainer(actor, reward_fn, torch.optim.AdamW(actor.parameters(), lr=5e-3),
                          group_size=4, kl_coef=0.0, ppo_epochs=2)
    prompts = torch.randint(0, 64, (8, 6))

    def low_frac():
        with torch.no_grad():
            fr = [float((trainer._rollout(prompts, 8)[:, 6:] < 32).float().mean()) for _ in range(4)]
        return sum(fr) / len(fr)

    before = low_frac()
    for _ in range(20):
        trainer.train_step(prompts, max_new_tokens=8)
    after = low_frac()
    assert after > max(before + 0.2, 0.8), f"GRPO did not shift: {before:.2f} -> {after:.2f}"
