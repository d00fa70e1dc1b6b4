"""Flagship benchmark — the BASELINE.json headline metric.

Measures GRPO default-config training throughput (episodes/sec; reference
≈1 s/episode on 1x A100-40G, README.md:35-36) on Qwen2.5-1.5B-Instruct
architecture with random-init weights, deberta-v3-large-shaped random-init
reward model, synthetic hh-rlhf-shaped prompts (no network for datasets/
checkpoints), response_length=1500, LoRA r=64, 512 prompts x N=4 samples
per update per rank (weak scaling: per-GPU work fixed as N grows).

One "step" = one full GRPO update: in-process rollout (paged-KV HIP
sampler) -> reward model pass -> scoring pass (policy+ref logprobs, fused
HIP logprob kernel) -> PPO minibatch update (HIP AdamW).

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
Multi-GPU: launched by the driver via torch.distributed.run, one rank per
GPU over RCCL; rank 0 prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--response-length", type=int, default=1500)
    ap.add_argument("--prompts-per-rank", type=int, default=512)
    ap.add_argument("--sample-n", type=int, default=4)
    # Defaults promoted after the 5-seed reward-curve validation
    # (profiles/rollout_modes_SUMMARY.md): fp8 e4m3 KV cache + the
    # sampler's own behavior-policy logprobs train indistinguishably from
    # the bf16/recompute path.  Compute dtype stays bf16 throughout.
    ap.add_argument("--kv-dtype", type=str, default="fp8_e4m3",
                    help="fp8_e4m3 (default) | bf16 (reference-faithful path)")
    ap.add_argument("--rollout-logprobs", dest="rollout_logprobs",
                    action="store_true", default=True,
                    help="use sampler-reported behavior logprobs (default)")
    ap.add_argument("--no-rollout-logprobs", dest="rollout_logprobs",
                    action="store_false",
                    help="recompute behavior logprobs in the scoring pass "
                         "(the reference's behavior)")
    ap.add_argument("--rollout-weight-dtype", type=str, default="bf16",
                    help="bf16 (default) | fp8_e4m3 (e4m3 merged rollout "
                         "weights; secondary measurement until promoted)")
    args = ap.parse_args()

    from nanorlhf_amd.algos import grpo
    from nanorlhf_amd.algos.grpo import GRPOConfig
    from nanorlhf_amd.data import hh_shaped_prompts
    from nanorlhf_amd.models import CausalLM, ScalarHeadModel, get_config
    from nanorlhf_amd.parallel import dist as pdist
    from nanorlhf_amd.rewards import ModelReward
    from nanorlhf_amd.utils.offload import OffloadEngine

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    assert torch.cuda.is_available(), "bench.py needs an MI355X"

    # GRPO default batch algebra: 4 * 8 * 16 = 512 prompts/update/rank
    # (reference GRPO/grpo_trainer.py:216-247)
    n_prompts = args.prompts_per_rank
    accum = 8
    num_mini = 16
    micro = max(1, n_prompts // (accum * num_mini))
    cfg = GRPOConfig(
        model_preset="qwen2.5-1.5b", dtype="bfloat16",
        use_lora=True, lora_r=64, lora_alpha=16,
        per_device_train_batch_size=micro, gradient_accumulation_steps=accum,
        num_mini_batches=num_mini, total_episodes=10**9,
        sample_n=args.sample_n, response_length=args.response_length,
        temperature=0.7, top_p=0.95, stop_token_id=1, pad_token_id=0,
        kl_coef=0.05, learning_rate=3e-6,
        # 288 GB HBM: no recompute needed at these batch shapes
        gradient_checkpointing=False,
        score_token_budget=131072,
        train_token_budget=49152,
        kv_cache_dtype=args.kv_dtype,
        rollout_weight_dtype=args.rollout_weight_dtype,
        use_rollout_logprobs=args.rollout_logprobs,
        output_dir=os.environ.get("BENCH_OUT", "/tmp/nanorlhf_bench"),
        save_steps=0, log_samples=0, report_to="none",
        missing_eos_penalty=1.0,
    )

    torch.manual_seed(1234)
    mcfg = get_config("qwen2.5-1.5b")
    # construct in bf16 directly: fp32 CPU init would transiently cost
    # ~14 GB host RAM per rank (x8 ranks in the scaling run)
    torch.set_default_dtype(torch.bfloat16)
    policy = CausalLM(mcfg)
    ref = CausalLM(mcfg)
    ref.load_state_dict(policy.state_dict())
    rm_model = ScalarHeadModel.from_preset("rm-large")
    torch.set_default_dtype(torch.float32)
    prompts = hh_shaped_prompts(2048, mcfg.vocab_size, seed=7)

    # trainer wires dist/devices itself (reads RANK/LOCAL_RANK/WORLD_SIZE)
    reward_stub = lambda seqs: torch.zeros(len(seqs))  # replaced below
    tr = grpo.make_trainer(cfg, policy, ref, reward_stub, prompts)
    rm_model = rm_model.to(tr.device).to(torch.bfloat16)
    tr.reward_fn = ModelReward(rm_model, tr.device, token_budget=65536)

    dev = tr.device
    def one_update(u):
        ro, gs = tr._rollout(u)
        with tr.timers.phase("score"):
            td = tr.algo.make_train_data(tr, ro, gs)
        with tr.timers.phase("update"):
            tr._update(td)

    # quiet the per-update metric machinery during timing
    for w in range(args.warmup):
        one_update(u=1000 + w)
    pdist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for s in range(args.steps):
        one_update(u=2000 + s)
    pdist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    if world > 1:
        import torch.distributed as dist_
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=dev if dist_.get_backend() == "nccl" else "cpu")
        dist_.all_reduce(t, op=dist_.ReduceOp.MAX)
        elapsed = float(t.item())

    local_batch = tr.sizes["local_batch_size"]  # actual prompts consumed/update
    episodes_total = local_batch * world * args.steps
    eps_per_sec = episodes_total / elapsed
    if rank == 0:
        phases = tr.timers.snapshot_and_reset()
        print(f"[bench] phase totals over {args.warmup}+{args.steps} updates: " +
              ", ".join(f"{k}={v:.1f}s" for k, v in sorted(phases.items())
                        if isinstance(v, float)), file=sys.stderr)
    ms_per_step = elapsed / args.steps * 1000.0
    if rank == 0:
        print(json.dumps({
            "metric": "episodes/sec (seconds/episode) Qwen2.5-1.5B GRPO default, 1/2/4/8 MI355X",
            "value": eps_per_sec,
            "unit": "episodes/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": eps_per_sec / 1.0,  # reference: ~1 episode/sec (1 s/episode, A100-40G)
            "dtype": "bf16",
            "kv_cache_dtype": args.kv_dtype,
            "data": "synthetic hh-rlhf-shaped prompts, random-init weights (no network)",
            "config": {
                "model": "qwen2.5-1.5b-instruct-arch (random init)",
                "reward_model": "deberta-v3-large-shaped encoder (random init)",
                "global_batch": local_batch * world,
                "sample_n": args.sample_n,
                "seq_len": args.response_length,
                "lora_r": 64,
                "parallelism": f"dp{world}",
            },
        }))


if __name__ == "__main__":
    main()
