"""PPO entry point with value model + value initializer (reference
PPO/ppo.py: separate policy/value LRs :118-119, value model loaded as a
scalar-head classifier :280-287, finetuned_value_model before training
:371-380)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from nanorlhf_amd.algos import ppo
from nanorlhf_amd.algos.ppo import PPOConfig
from nanorlhf_amd.algos.value_init import finetune_value_model
from nanorlhf_amd.data import hh_shaped_prompts
from nanorlhf_amd.models import CausalLM, ScalarHeadModel, get_config
from nanorlhf_amd.rewards import ModelReward

ON_GPU = torch.cuda.is_available()
config = PPOConfig(
    exp_name="ppo_default",
    output_dir="runs/ppo",
    seed=2434,
    model_preset="qwen2.5-1.5b" if ON_GPU else "tiny",
    reward_preset="rm-large" if ON_GPU else "rm-tiny",
    dtype="bfloat16" if ON_GPU else "float32",
    use_lora=True, lora_r=64 if ON_GPU else 4, lora_alpha=16,
    per_device_train_batch_size=4 if ON_GPU else 2,
    gradient_accumulation_steps=8 if ON_GPU else 2,
    num_mini_batches=8 if ON_GPU else 2,     # PPO token budget 16*2316 (ppo_trainer.py:594)
    total_episodes=100_000,
    sample_n=1, response_length=1500 if ON_GPU else 16,
    temperature=0.7, top_p=0.95, stop_token_id=1,
    kl_coef=0.05, cliprange=0.2, cliprange_value=0.2, vf_coef=0.1,
    gamma=1.0, lam=0.95,                      # GAE (ppo_trainer.py:688-697)
    learning_rate=3e-6, value_learning_rate=3e-6,  # ppo.py:118-119
    score_token_budget=16 * 2316,
    save_steps=1,
    missing_eos_penalty=1.0,
)

if __name__ == "__main__":
    mcfg = get_config(config.model_preset)
    policy = CausalLM(mcfg)
    ref_policy = CausalLM(mcfg)
    ref_policy.load_state_dict(policy.state_dict())
    # critic: causal scalar-head model (AutoModelForSequenceClassification
    # num_labels=1 role, ppo.py:280-287)
    value_model = ScalarHeadModel.from_preset(config.model_preset, num_labels=1,
                                              bidirectional=False)
    device = torch.device("cuda:0" if ON_GPU else "cpu")
    rm = ScalarHeadModel.from_preset(config.reward_preset)
    if ON_GPU:
        rm = rm.to(device).to(torch.bfloat16)
    reward_fn = ModelReward(rm, device)
    prompts = hh_shaped_prompts(2048, mcfg.vocab_size, seed=0)

    trainer = ppo.make_trainer(config, policy, ref_policy, reward_fn, prompts,
                               value_model=value_model)
    # value initializer: one rollout batch of 500 prompts, regression on
    # KL-shaped returns, early stop (PPO/value_initializer.py:69-388)
    stats = finetune_value_model(trainer, num_prompts=500 if ON_GPU else 8,
                                 epochs=8, lr=1e-5)
    print(f"value init: {stats['epochs_ran']} epochs, "
          f"best eval loss {stats['best_eval_loss']:.4f}")
    trainer.train(num_updates=3)
    trainer.save()
