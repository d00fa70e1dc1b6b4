"""All-in-one training config.

The reference keeps every knob in a per-algorithm dataclass literal inside
the entry script ("ALL setting is on the file you run", README.md:34;
GRPO/grpo.py:86-155).  We keep that user surface: one dataclass, per-algo
subclasses with their extra fields, instantiated with literals in
examples/<algo>.py.  Batch-size algebra fields keep trl/reference names and
semantics (grpo_trainer.py:216-247) so configs transfer."""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class RLHFConfig:
    exp_name: str = "nanorlhf"
    output_dir: str = "runs/default"
    seed: int = 2434

    # ---- models ------------------------------------------------------------
    model_preset: str = "qwen2.5-1.5b"
    reward_preset: str | None = "rm-large"
    dtype: str = "bfloat16"

    # ---- LoRA (reference defaults grpo.py:90-99,226-243) --------------------
    use_lora: bool = True
    lora_r: int = 64
    lora_alpha: int = 16
    lora_dropout: float = 0.0

    # ---- batch algebra (trl names; grpo_trainer.py:216-247) -----------------
    per_device_train_batch_size: int = 4      # micro-batch rows
    gradient_accumulation_steps: int = 8
    num_mini_batches: int = 16
    total_episodes: int = 100_000
    num_ppo_epochs: int = 1

    # ---- rollout -------------------------------------------------------------
    sample_n: int = 1                 # samples per prompt (GRPO/RLOO N, RAFT K)
    response_length: int = 1500
    temperature: float = 0.7
    top_p: float = 0.95
    stop_token_id: int | None = None  # EOS; entry scripts set it
    pad_token_id: int = 0
    reseed_rollouts: bool = True      # new sampler seed per update (ref :127)
    # use the sampler's own per-token logprobs as the behavior-policy
    # ("old") logprobs, skipping the policy half of the scoring pass.  The
    # reference recomputes them (its vLLM flow discards logprobs); default
    # keeps that behavior, the flag is the in-process-sampler shortcut.
    use_rollout_logprobs: bool = False

    # ---- objective ----------------------------------------------------------
    kl_coef: float = 0.05
    cliprange: float = 0.2
    gamma: float = 1.0
    lam: float = 0.95                 # PPO GAE lambda
    vf_coef: float = 0.1
    cliprange_value: float = 0.2
    advantage_whiten: bool = False    # REINFORCE defaults True (reinforce.py:103)
    whiten_rewards: bool = False
    missing_eos_penalty: float | None = 1.0

    # ---- optimizer / schedule ----------------------------------------------
    learning_rate: float = 3e-6
    min_lr_ratio: float = 0.1          # cosine_with_min_lr (grpo.py:119-120)
    lr_scheduler_type: str = "cosine_with_min_lr"  # or "constant", "reduce_lr_on_plateau"
    plateau_patience: int = 10       # reduce_lr_on_plateau (PPO/ppo.py:97-98)
    plateau_factor: float = 0.5
    warmup_steps: int = 0
    weight_decay: float = 0.0
    adam_beta1: float = 0.9
    adam_beta2: float = 0.95
    adam_eps: float = 1e-8
    max_grad_norm: float | None = None
    gradient_checkpointing: bool = True
    # reference quirk: optimizer.step() every micro-batch inside accumulate()
    # (grpo_trainer.py:692).  Default FIXED (step at accumulation boundary);
    # set True to reproduce the reference exactly.
    step_every_microbatch: bool = False

    # ---- memory / scheduling budgets ----------------------------------------
    score_token_budget: int = 22 * 2316    # fwd-scoring bucket budget (ref :534)
    train_token_budget: int = 0            # 0 → derive from micro-batch rows
    kv_pool_tokens: int = 0                # 0 → auto from batch & lengths
    kv_cache_dtype: str = "bf16"           # "fp8_e4m3": OCP fp8 paged KV (halves
                                           # the decode KV stream; opt-in)
    rollout_weight_dtype: str = "bf16"     # "fp8_e4m3": e4m3 MERGED rollout
                                           # weights via hipBLASLt fp8 GEMMs
                                           # (training weights untouched)
    offload_ref: bool | None = None        # None → auto by memory pressure
    offload_reward: bool | None = None
    offload_optimizer: bool = False

    # ---- sparse-GRPO / r1 mode (grpo_r1_trainer.py) --------------------------
    sparse_filter: bool = False            # drop score==0 samples (:565-568)

    # ---- DP load balance ------------------------------------------------------
    # Real-model rollouts have EOS-driven response-length variance; ranks
    # then carry unequal token counts into the scoring/update phases (the
    # synthetic bench is balanced).  When on, whole sample-groups are
    # re-assigned across ranks after reward so per-rank token totals are
    # near-equal (LPT greedy, deterministic).  Off by default: it is an
    # opt-in for real-data DP runs (SURVEY §7 hard part (d)).
    dp_rebalance_rollout: bool = False

    # ---- eval (r1 mode: greedy accuracy pass, grpo_r1_trainer.py:824-825) ----
    eval_steps: int = 0                    # 0 → no periodic eval
    eval_at_start: bool = True             # initial accuracy (:471-473)

    # ---- checkpoint / logging ------------------------------------------------
    save_steps: int = 0                    # 0 → no periodic save (ref default 1)
    save_total_limit: int = 8
    metric_for_best_model: str | None = "eval_objective/rlhf_reward_old"
    report_to: str = "none"                # "wandb" | "none"
    log_samples: int = 5

    # ------------------------------------------------------------------ algebra
    def batch_sizes(self, world_size: int) -> dict:
        local_batch_size = (self.per_device_train_batch_size
                            * self.gradient_accumulation_steps
                            * self.num_mini_batches)
        batch_size = local_batch_size * world_size
        local_mini_batch_size = local_batch_size // self.num_mini_batches
        num_updates = max(1, self.total_episodes // batch_size)
        return dict(local_batch_size=local_batch_size,
                    batch_size=batch_size,
                    local_mini_batch_size=local_mini_batch_size,
                    micro_batch_size=self.per_device_train_batch_size * world_size,
                    num_updates=num_updates)
