"""RLOO entry point (reference RLOO/rloo.py: rloo_sample_N=4 :107,
leave-one-out baseline, sequence-level clipped ratio)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from nanorlhf_amd.algos import rloo
from nanorlhf_amd.algos.rloo import RLOOConfig
from nanorlhf_amd.data import hh_shaped_prompts
from nanorlhf_amd.models import CausalLM, ScalarHeadModel, get_config
from nanorlhf_amd.rewards import ModelReward

ON_GPU = torch.cuda.is_available()
config = RLOOConfig(
    exp_name="rloo_default", output_dir="runs/rloo", seed=2434,
    # BASELINE config #4 runs this at qwen2.5-7b, DP=8 (torchrun, one rank/GPU)
    model_preset="qwen2.5-1.5b" if ON_GPU else "tiny",
    reward_preset="rm-large" if ON_GPU else "rm-tiny",
    dtype="bfloat16" if ON_GPU else "float32",
    use_lora=True, lora_r=64 if ON_GPU else 4, lora_alpha=16,
    per_device_train_batch_size=4 if ON_GPU else 2,
    gradient_accumulation_steps=8 if ON_GPU else 2,
    num_mini_batches=16 if ON_GPU else 2,
    total_episodes=100_000,
    sample_n=4, response_length=1500 if ON_GPU else 16,
    temperature=0.7, top_p=0.95, stop_token_id=1,
    kl_coef=0.05, cliprange=0.2, missing_eos_penalty=1.0,
    learning_rate=3e-6, save_steps=1,
)

if __name__ == "__main__":
    mcfg = get_config(config.model_preset)
    policy = CausalLM(mcfg)
    ref_policy = CausalLM(mcfg)
    ref_policy.load_state_dict(policy.state_dict())
    device = torch.device("cuda:0" if ON_GPU else "cpu")
    rm = ScalarHeadModel.from_preset(config.reward_preset)
    if ON_GPU:
        rm = rm.to(device).to(torch.bfloat16)
    trainer = rloo.make_trainer(config, policy, ref_policy, ModelReward(rm, device),
                                hh_shaped_prompts(2048, mcfg.vocab_size, seed=0))
    trainer.train(num_updates=3)
    trainer.save()
