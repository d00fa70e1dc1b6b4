"""Sparse-GRPO "r1" entry point — the R1-Zero reproduction mode
(reference examples/r1-v0/grpo_r1.py + grpo_r1_trainer.py):

  * base (non-instruct) policy, long responses (8000 tokens, :145)
  * rule-based math reward: boxed-answer extraction + 3-way equivalence
    under a subprocess timeout (:179-273)
  * sparse GRPO: drop zero-advantage samples (:565-568)
  * dynamic length-bucketed mini-batching by token budget (:410-435,
    :589 rollout budget, :700 train budget)
  * greedy eval-accuracy pass before training and every 10 steps
    (:471-473, :824-825)

With no network this runs a synthetic MetaMathQA-shaped task: prompts are
random token sequences whose gold answer is a deterministic function of the
prompt; responses are detokenized to digit strings so the rule-reward path
(extraction → normalization → sympy equivalence w/ timeout) is exercised
end to end."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from nanorlhf_amd.algos import grpo
from nanorlhf_amd.algos.grpo import GRPOConfig
from nanorlhf_amd.data import math_shaped_prompts
from nanorlhf_amd.models import CausalLM, get_config
from nanorlhf_amd.rewards import MathRuleReward
from nanorlhf_amd.sampler import SamplingParams

ON_GPU = torch.cuda.is_available()
config = GRPOConfig(
    exp_name="grpo_r1", output_dir="runs/grpo_r1", seed=2434,
    model_preset="qwen2.5-1.5b" if ON_GPU else "tiny",  # base model (grpo_r1.py:92)
    dtype="bfloat16" if ON_GPU else "float32",
    use_lora=True, lora_r=64 if ON_GPU else 4, lora_alpha=16,
    per_device_train_batch_size=4 if ON_GPU else 2,
    gradient_accumulation_steps=8 if ON_GPU else 2,
    num_mini_batches=16 if ON_GPU else 2,
    total_episodes=100_000,
    sample_n=4,
    response_length=8000 if ON_GPU else 16,   # grpo_r1.py:145
    temperature=0.7, top_p=0.95, stop_token_id=1,
    kl_coef=0.0,                              # grpo_r1.py:138-145
    cliprange=0.2,
    sparse_filter=True,                       # sparse GRPO
    score_token_budget=22 * 2316,             # rollout-phase bucket budget (:589)
    train_token_budget=4 * 2316,              # train-phase bucket budget (:700)
    missing_eos_penalty=None,
    eval_steps=10, eval_at_start=True,
    learning_rate=3e-6, save_steps=10,
)


def detokenize(ids):
    """Synthetic detokenizer: ids → digit words (rule reward parses text)."""
    return " ".join(str(t) for t in ids)


def gold_answer_for(prompt):
    """Synthetic MetaMathQA-style answer map (reference hashes the train
    set's questions → answers, grpo_r1.py:237-240)."""
    return str(sum(prompt) % 1000)


def make_eval_fn(eval_prompts, gold):
    def eval_fn(trainer):
        """Greedy MATH-500-style accuracy pass (grpo_r1.py:276-341)."""
        params = SamplingParams(n=1, temperature=0.0, top_p=1.0,
                                max_tokens=min(trainer.cfg.response_length, 512),
                                seed=0, stop_token_id=trainer.cfg.stop_token_id)
        out = trainer.sampler.generate(eval_prompts, params,
                                       pad_token_id=trainer.cfg.pad_token_id)
        texts = [detokenize([t for t in row if t != trainer.cfg.pad_token_id])
                 for row in out.tolist()]
        scores = MathRuleReward(gold, require_boxed=False)(texts)
        lens = [(row != trainer.cfg.pad_token_id).sum() for row in out]
        return {"accuracy": float(scores.mean()),
                "response_length": float(torch.tensor([float(l) for l in lens]).mean())}
    return eval_fn


# Real-data seam (reference grpo_r1.py:92,126-128,237-273): point MODEL_PATH
# at a local HF Qwen2 checkpoint and DATA_JSONL at a jsonl of
# {"question": ..., "answer": ...} records (MetaMathQA-style) to train on
# real text with the rule reward over decoded strings.
MODEL_PATH = None        # e.g. "/models/Qwen2-1.5B"
DATA_JSONL = None        # e.g. "data/metamathqa.jsonl"

def run_real(model_path: str, data_jsonl: str, cfg=None, num_updates=None):
    """Real-data r1 training: HF checkpoint + tokenizer + (question, answer)
    jsonl records with the rule reward over decoded strings (reference
    grpo_r1.py:92,126-128,237-273)."""
    import json as _json

    from nanorlhf_amd.data.tokenizer import load_tokenizer, prepare_math_prompts
    from nanorlhf_amd.models.hf_import import load_pretrained
    from nanorlhf_amd.rewards import StringReward
    from nanorlhf_amd.rewards.mathcheck import answers_equal, extract_boxed

    cfg = cfg or config
    policy = load_pretrained(model_path)
    ref_policy = load_pretrained(model_path)
    tokenizer = load_tokenizer(model_path)
    cfg.stop_token_id = tokenizer.eos_token_id
    cfg.pad_token_id = tokenizer.pad_token_id
    with open(data_jsonl) as f:
        records = [_json.loads(line) for line in f]
    train_prompts, answer_map = prepare_math_prompts(records, tokenizer,
                                                     max_prompt_len=512)

    def rule_reward(texts, responses_ids, tok):
        """Reference r1 contract (grpo_r1.py:250-273): boxed-answer
        extraction + structured equivalence, gold looked up by the
        prompt prefix of the decoded string."""
        out = []
        for t in texts:
            gold = next((a for q, a in answer_map.items()
                         if t.startswith(q)), None)
            pred = extract_boxed(t)
            ok = (gold is not None and pred is not None
                  and answers_equal(pred, gold, sympy_timeout_s=0.015))
            out.append(1.0 if ok else 0.0)
        return torch.tensor(out)

    reward_fn = StringReward(rule_reward, tokenizer, mode="r1")
    trainer = grpo.make_trainer(cfg, policy, ref_policy, reward_fn,
                                train_prompts)
    trainer.train(num_updates=num_updates)
    trainer.save()
    return trainer


if __name__ == "__main__":
    if MODEL_PATH is not None and DATA_JSONL is not None:
        run_real(MODEL_PATH, DATA_JSONL)
        sys.exit(0)

    mcfg = get_config(config.model_preset)
    policy = CausalLM(mcfg)
    ref_policy = CausalLM(mcfg)
    ref_policy.load_state_dict(policy.state_dict())

    train_prompts = math_shaped_prompts(1024 if ON_GPU else 64, mcfg.vocab_size, seed=0)
    eval_prompts = math_shaped_prompts(64 if ON_GPU else 8, mcfg.vocab_size, seed=99)
    train_gold = [gold_answer_for(p) for p in train_prompts]
    eval_gold = [gold_answer_for(p) for p in eval_prompts]

    # rule reward over detokenized prompt+response text, gold looked up by
    # rollout order (trainer passes prompt+response id sequences)
    prompt_index = {tuple(p): i for i, p in enumerate(train_prompts)}

    def reward_fn(sequences):
        texts, keys = [], []
        for seq in sequences:
            # split prompt from response by matching known prompts
            gold_i = None
            for plen in range(min(len(seq), 512), 0, -1):
                gi = prompt_index.get(tuple(seq[:plen]))
                if gi is not None:
                    gold_i = gi
                    break
            texts.append(detokenize(seq[plen:] if gold_i is not None else seq))
            keys.append(gold_i)
        golds = [train_gold[k] if k is not None else "" for k in keys]
        return MathRuleReward(golds, require_boxed=False)(texts)

    trainer = grpo.make_trainer(config, policy, ref_policy, reward_fn, train_prompts)
    trainer.eval_fn = make_eval_fn(eval_prompts, eval_gold)
    trainer.train(num_updates=3)
    trainer.save()
