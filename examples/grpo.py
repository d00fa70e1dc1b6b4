"""GRPO entry point — ALL settings live in this file (the reference's user
surface: "ALL setting is on the file you run", README.md:34; config fields
mirror GRPO/grpo.py:86-155).

Run:  python examples/grpo.py

With no network access this demonstrates the full pipeline on synthetic
hh-rlhf-shaped prompts with a random-init policy + reward model.  Point
`model_preset` / load real weights (models/config.py presets match
Qwen2.5-{0.5,1.5,7}B shapes) for real runs."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from nanorlhf_amd.algos import grpo
from nanorlhf_amd.algos.grpo import GRPOConfig
from nanorlhf_amd.data import hh_shaped_prompts
from nanorlhf_amd.models import CausalLM, ScalarHeadModel, get_config
from nanorlhf_amd.rewards import ModelReward

# --------------------------------------------------------------------------
# config — every knob in one place (reference grpo.py:108-155)
# --------------------------------------------------------------------------
ON_GPU = torch.cuda.is_available()
config = GRPOConfig(
    exp_name="grpo_default",
    output_dir="runs/grpo",
    seed=2434,                                 # grpo.py:77-78
    model_preset="qwen2.5-1.5b" if ON_GPU else "tiny",
    reward_preset="rm-large" if ON_GPU else "rm-tiny",
    dtype="bfloat16" if ON_GPU else "float32",
    # LoRA (grpo.py:90-99): r=64, alpha=16, embed/lm_head fully trained
    use_lora=True, lora_r=64 if ON_GPU else 4, lora_alpha=16,
    # batch algebra (grpo_trainer.py:216-247): 4 * 8 * 16 = 512 prompts/update
    per_device_train_batch_size=4 if ON_GPU else 2,
    gradient_accumulation_steps=8 if ON_GPU else 2,
    num_mini_batches=16 if ON_GPU else 2,
    total_episodes=100_000,
    # rollout (grpo.py:106, SamplingParams at grpo_trainer.py:127)
    sample_n=4, response_length=1500 if ON_GPU else 16,
    temperature=0.7, top_p=0.95, stop_token_id=1,
    # objective
    kl_coef=0.05, cliprange=0.2, advantage_whiten=False,  # grpo.py:104
    missing_eos_penalty=1.0,
    # optimizer (grpo.py:119-120 cosine_with_min_lr)
    learning_rate=3e-6, lr_scheduler_type="cosine_with_min_lr", min_lr_ratio=0.1,
    # checkpoints (grpo.py:143,150): save every step, keep 8, best by *_old
    save_steps=1, save_total_limit=8,
    metric_for_best_model="eval_objective/rlhf_reward_old",
    report_to="none",                           # set "wandb" + WANDB_PROJECT
)

# Point this at a local HF Qwen2.5 checkpoint directory (config.json +
# safetensors + tokenizer files) to train REAL weights on REAL text
# (reference grpo.py:209-270).  None → synthetic demo (no network here).
MODEL_PATH = None        # e.g. "/models/Qwen2.5-1.5B-Instruct"
RM_PATH = None           # sequence-classifier reward checkpoint (optional)

if __name__ == "__main__":
    tokenizer = None
    if MODEL_PATH is not None:
        # real-model seam: HF safetensors -> fused layout + tokenizer
        from nanorlhf_amd.data.tokenizer import load_tokenizer, prepare_hh_prompts
        from nanorlhf_amd.models.hf_import import load_pretrained
        policy = load_pretrained(MODEL_PATH)
        ref_policy = load_pretrained(MODEL_PATH)
        tokenizer = load_tokenizer(MODEL_PATH)
        config.stop_token_id = tokenizer.eos_token_id
        config.pad_token_id = tokenizer.pad_token_id
        mcfg = policy.cfg
    else:
        mcfg = get_config(config.model_preset)
        policy = CausalLM(mcfg)
        ref_policy = CausalLM(mcfg)
        ref_policy.load_state_dict(policy.state_dict())

    # model-based reward (grpo.py:162-198: deberta-v3-large RM shuttled
    # on/off GPU → here an OffloadEngine policy inside ModelReward)
    device = torch.device("cuda:0" if ON_GPU else "cpu")
    rm = ScalarHeadModel.from_preset(config.reward_preset)
    if ON_GPU:
        rm = rm.to(device).to(torch.bfloat16)
    # host-offload policy for the RM (the reference shuttles it CPU<->GPU per
    # reward pass, grpo.py:164,195; on 288 GB the auto policy keeps it
    # resident and only offloads under real pressure)
    from nanorlhf_amd.utils.offload import OffloadEngine
    reward_fn = ModelReward(
        rm, device,
        offload=OffloadEngine(device, enabled=config.offload_reward) if ON_GPU else None)

    if tokenizer is not None:
        # real text path: hh-rlhf-style records through the chat scaffold
        # (grpo.py:249-270).  Reads a local jsonl of {"chosen": transcript}
        # records; reward over decoded strings via StringReward when a
        # string reward_func is preferred over the id-level ModelReward.
        import json as _json
        DATA_JSONL = "data/hh_train.jsonl"
        with open(DATA_JSONL) as f:
            records = [_json.loads(line) for line in f]
        prompts = prepare_hh_prompts(records, tokenizer, max_prompt_len=512)
    else:
        # synthetic hh-rlhf-shaped prompts (grpo.py:249-270 prompt prep)
        prompts = hh_shaped_prompts(2048, mcfg.vocab_size, seed=0)

    # early stopping wired like the reference (patience 10^6 — effectively
    # disabled; grpo.py:89,281)
    from nanorlhf_amd.utils.callbacks import EarlyStoppingCallback
    trainer = grpo.make_trainer(
        config, policy, ref_policy, reward_fn, prompts,
        callbacks=[EarlyStoppingCallback(
            metric="eval_objective/rlhf_reward_old", patience=10**6)])
    trainer.train(num_updates=3)
    trainer.save()
