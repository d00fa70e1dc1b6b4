"""Shared RLHF trainer engine.

One engine; each algorithm (GRPO/PPO/RLOO/ReMax/RAFT/REINFORCE) reduces to
an AlgoSpec plug-in providing its advantage construction and loss — exactly
the delta SURVEY.md §2.1 shows between the reference's six forked 700-800
line trainer files.  Per-update pipeline mirrors the reference's train()
(GRPO/grpo_trainer.py:406-778):

  rollout (in-process sampler) → reward → [algo row selection] → scoring
  pass (policy+ref logprobs, fused HIP logprob kernel, token-budget buckets)
  → advantages → minibatch PPO update → metrics/checkpoint.

Differences by design (MI355X-first): no disk round trip for generation, no
CPU shuttles unless memory pressure demands (utils/offload.py), packed
varlen forwards with dynamic token-budget bucketing everywhere (the r1
variant's throughput trick, grpo_r1_trainer.py:410-435, made default).
"""
from __future__ import annotations

import math
import time
from dataclasses import dataclass, field
from typing import Callable, Optional

import torch

from .. import ops
from ..config import RLHFConfig
from ..data.buckets import create_batches
from ..models.lora import LoraConfig, apply_lora
from ..models.qwen2 import AttnContext, CausalLM
from ..models.value_head import ScalarHeadModel
from ..parallel import dist as pdist
from ..parallel.ddp import GradReducer
from ..sampler.engine import SamplerEngine, SamplingParams
from ..utils.checkpoint import CheckpointManager
from ..utils.logging import MetricsLogger
from ..utils.offload import OffloadEngine
from ..utils.seed import rank_seed, set_seed
from ..utils.timers import PhaseTimers
from . import functional as F


@dataclass
class Rollout:
    """One update's sampled data (all lists indexed by row = prompt*N + j)."""
    prompts: list[list[int]]          # per row (prompt repeated N times)
    responses: list[list[int]]        # truncated at stop token
    scores: torch.Tensor              # [R] fp32 (after penalties)
    raw_scores: torch.Tensor          # [R] before penalties
    contains_eos: torch.Tensor        # [R] bool
    sample_n: int
    logprobs: Optional[list] = None   # sampler-reported per-token logprobs

    @property
    def num_rows(self):
        return len(self.prompts)

    def seq_lens(self) -> list[int]:
        return [len(p) + len(r) for p, r in zip(self.prompts, self.responses)]


@dataclass
class TrainData:
    """What the update loop consumes (row-indexed into its own tensors)."""
    rows: list[int]                       # rollout row ids being trained
    prompts: list[list[int]]
    responses: list[list[int]]
    old_logprobs: torch.Tensor            # [R, L] padded (INVALID_LOGPROB pads)
    ref_logprobs: Optional[torch.Tensor]  # [R, L]
    mask: torch.Tensor                    # [R, L] response-token mask
    advantages: torch.Tensor              # [R, L] token-level or [R] sequence-level
    sequence_level: bool = False
    values: Optional[torch.Tensor] = None     # [R, L] PPO
    returns: Optional[torch.Tensor] = None    # [R, L] PPO
    stats: dict = field(default_factory=dict)  # metric scalars from prep


class AlgoSpec:
    """Base algorithm plug-in. Subclasses override make_train_data and loss."""

    name = "base"
    needs_ref = True
    needs_value = False
    greedy_baseline = False   # ReMax second pass

    def make_train_data(self, trainer: "RLHFTrainer", ro: Rollout,
                        greedy_scores: Optional[torch.Tensor]) -> TrainData:
        raise NotImplementedError

    def loss(self, trainer: "RLHFTrainer", td: TrainData, mb: dict,
             new_logprobs: torch.Tensor, vpred: Optional[torch.Tensor]) -> tuple[torch.Tensor, dict]:
        """Default: token-level PPO-clip on td.advantages."""
        cfg = trainer.cfg
        pg_loss, st = F.ppo_clip_token_loss(new_logprobs, mb["old_logprobs"],
                                            mb["advantages"], mb["mask"], cfg.cliprange)
        return pg_loss, st


class RLHFTrainer:
    def __init__(self, cfg: RLHFConfig, algo: AlgoSpec, policy: CausalLM,
                 ref_policy: Optional[CausalLM], reward_fn: Callable,
                 train_prompts: list[list[int]],
                 value_model: Optional[ScalarHeadModel] = None,
                 device: Optional[torch.device] = None,
                 callbacks: Optional[list] = None):
        self.cfg = cfg
        self.algo = algo
        self.rank, self.local_rank, self.world = pdist.init_distributed()
        if device is None:
            device = torch.device(f"cuda:{self.local_rank}") if torch.cuda.is_available() \
                else torch.device("cpu")
        self.device = device
        set_seed(rank_seed(cfg.seed, self.rank))

        self.sizes = cfg.batch_sizes(self.world)
        F.exact_div(self.sizes["local_batch_size"], cfg.num_mini_batches,
                    "local_batch_size must divide num_mini_batches")

        dtype = getattr(torch, cfg.dtype)
        self.policy = policy.to(device)
        if device.type == "cuda":
            self.policy = self.policy.to(dtype)
        if cfg.use_lora:
            apply_lora(self.policy, LoraConfig(cfg.lora_r, cfg.lora_alpha, cfg.lora_dropout))
        self.policy.gradient_checkpointing = cfg.gradient_checkpointing
        if ref_policy is None and algo.needs_ref:
            raise ValueError(f"{algo.name} needs a reference policy")
        self.ref_policy = ref_policy.to(device).eval() if ref_policy is not None else None
        if self.ref_policy is not None and device.type == "cuda":
            self.ref_policy = self.ref_policy.to(dtype)
        for p in (self.ref_policy.parameters() if self.ref_policy is not None else []):
            p.requires_grad_(False)
        self.value_model = value_model.to(device) if value_model is not None else None
        if self.value_model is not None and device.type == "cuda":
            self.value_model = self.value_model.to(dtype)
        self.reward_fn = reward_fn
        self.offload = OffloadEngine(device, enabled=cfg.offload_ref)

        # sampler over the live policy
        pool_tokens = cfg.kv_pool_tokens or self._auto_pool_tokens(train_prompts)
        self.sampler = SamplerEngine(
            self.policy, kv_pool_tokens=pool_tokens,
            kv_cache_dtype=getattr(cfg, "kv_cache_dtype", "bf16"),
            rollout_weight_dtype=getattr(cfg, "rollout_weight_dtype", "bf16"))

        # optimizer over trainable params (+ value model for PPO joint update).
        # Reference parity (PPO/ppo_trainer.py:341-402): 4 groups —
        # policy/value × decay/no-decay (biases + norm weights never decay).
        params = [p for p in self.policy.parameters() if p.requires_grad]
        self.value_params = []
        if self.value_model is not None:
            self.value_params = [p for p in self.value_model.parameters() if p.requires_grad]
        groups = []
        for model, tag in ((self.policy, "policy"), (self.value_model, "value")):
            if model is None:
                continue
            decay, nodecay = _decay_split(model)
            groups.append({"params": decay, "weight_decay": cfg.weight_decay,
                           "name": tag + "_decay"})
            groups.append({"params": nodecay, "weight_decay": 0.0,
                           "name": tag + "_nodecay"})
        self.optimizer = ops.FusedAdamW(
            groups, lr=cfg.learning_rate, betas=(cfg.adam_beta1, cfg.adam_beta2),
            eps=cfg.adam_eps, weight_decay=cfg.weight_decay)
        for g in self.optimizer.param_groups:
            g.setdefault("initial_lr", cfg.learning_rate)
        self.lr_step = 0
        self.reducer = GradReducer(params + self.value_params)
        if self.world > 1:
            # ranks seed differently (rank_seed) BEFORE LoRA init → sync the
            # initial trainable state from rank 0, as DDP's broadcast does
            import torch.distributed as _dist
            for p in params + self.value_params:
                _dist.broadcast(p.data, 0)

        self.train_prompts = train_prompts
        self._prompt_cursor = 0
        g = torch.Generator().manual_seed(rank_seed(cfg.seed, self.rank))
        self._shuffle = torch.randperm(len(train_prompts), generator=g).tolist()
        self._keep_gen = g

        self.ckpt = CheckpointManager(cfg.output_dir, cfg.save_total_limit,
                                      cfg.metric_for_best_model)
        self.logger = MetricsLogger(cfg.output_dir, cfg.report_to, rank=self.rank)
        self.timers = PhaseTimers(sync_cuda=device.type == "cuda")
        self.global_step = 0
        self.episode = 0
        # optional evaluation hook: eval_fn(trainer) -> dict of metrics
        # (r1 mode's greedy MATH accuracy pass, grpo_r1_trainer.py:471-473,824-825)
        self.eval_fn = None
        self.callbacks = list(callbacks or [])
        self._plateau_best = None
        self._plateau_bad = 0
        self._lr_scale = 1.0

    # ------------------------------------------------------------------ utils
    def _auto_pool_tokens(self, prompts) -> int:
        from ..sampler.cache import PagedKVCache
        from ..sampler.engine import SamplerEngine
        max_prompt = max((len(p) for p in prompts), default=256)
        per_seq = max_prompt + self.cfg.response_length
        n_seq = self.sizes["local_batch_size"] * max(self.cfg.sample_n, 1)
        want = per_seq * min(n_seq, 4096) + 4096
        if self.device.type == "cuda":
            # cap the pool against free HBM (long-response configs like r1's
            # 8000 tokens would otherwise over-allocate); continuous batching
            # queues the sequences that don't fit concurrently
            kv_dtype = SamplerEngine.KV_DTYPES.get(
                getattr(self.cfg, "kv_cache_dtype", "bf16"), torch.bfloat16)
            bpt = PagedKVCache.bytes_per_token(
                self.policy.cfg, dtype=kv_dtype)
            free, _ = torch.cuda.mem_get_info(self.device)
            cap = int(free * 0.55) // bpt
            if want > cap:
                want = max(cap, per_seq * 8)  # keep at least a few sequences
        return want

    def _next_prompts(self) -> list[list[int]]:
        n = self.sizes["local_batch_size"]
        out = []
        while len(out) < n:
            if self._prompt_cursor >= len(self._shuffle):
                self._prompt_cursor = 0
                self._shuffle = torch.randperm(len(self.train_prompts),
                                               generator=self._keep_gen).tolist()
            out.append(self.train_prompts[self._shuffle[self._prompt_cursor]])
            self._prompt_cursor += 1
        return out

    def _lr(self) -> float:
        cfg = self.cfg
        t = self.lr_step
        total = self.sizes["num_updates"] * cfg.num_ppo_epochs * cfg.num_mini_batches
        if cfg.warmup_steps and t < cfg.warmup_steps:
            return cfg.learning_rate * (t + 1) / cfg.warmup_steps
        if cfg.lr_scheduler_type == "constant":
            return cfg.learning_rate
        if cfg.lr_scheduler_type == "reduce_lr_on_plateau":
            # reference PPO value path (ppo.py:97-98): halve on plateau
            return cfg.learning_rate * self._lr_scale
        # cosine_with_min_lr (reference grpo.py:119-120)
        min_lr = cfg.learning_rate * cfg.min_lr_ratio
        prog = min(1.0, t / max(1, total))
        return min_lr + 0.5 * (cfg.learning_rate - min_lr) * (1 + math.cos(math.pi * prog))

    # --------------------------------------------------------------- rollout
    def _rollout(self, update_idx: int) -> tuple[Rollout, Optional[torch.Tensor]]:
        cfg = self.cfg
        prompts = self._next_prompts()
        seed = rank_seed(cfg.seed, self.rank) + (update_idx * 7919 if cfg.reseed_rollouts else 0)
        params = SamplingParams(n=cfg.sample_n, temperature=cfg.temperature,
                                top_p=cfg.top_p, max_tokens=cfg.response_length,
                                seed=seed, stop_token_id=cfg.stop_token_id)
        with self.timers.phase("rollout"):
            resp_pad, lp_pad = self.sampler.generate(
                prompts, params, pad_token_id=cfg.pad_token_id, return_logprobs=True)
        greedy_scores = None
        if self.algo.greedy_baseline:
            gparams = SamplingParams(n=1, temperature=0.0, top_p=1.0,
                                     max_tokens=cfg.response_length, seed=seed,
                                     stop_token_id=cfg.stop_token_id)
            with self.timers.phase("rollout_greedy"):
                greedy_pad = self.sampler.generate(prompts, gparams,
                                                   pad_token_id=cfg.pad_token_id)


        def depad(resp_rows, n):
            rows_p, rows_r = [], []
            for i, row in enumerate(resp_rows.tolist()):
                prompt = prompts[i // n]
                if cfg.stop_token_id is not None and cfg.stop_token_id in row:
                    row = row[: row.index(cfg.stop_token_id) + 1]
                else:
                    while row and row[-1] == cfg.pad_token_id:
                        row.pop()
                rows_p.append(list(prompt))
                rows_r.append(row)
            return rows_p, rows_r

        rows_p, rows_r = depad(resp_pad, cfg.sample_n)
        rows_lp = [lp_pad[i, : len(r)].tolist() for i, r in enumerate(rows_r)]
        with self.timers.phase("reward"):
            raw_scores = self._call_reward(rows_p, rows_r)
            if self.algo.greedy_baseline:
                gp, gr = depad(greedy_pad if not isinstance(greedy_pad, tuple)
                               else greedy_pad[0], 1)
                greedy_scores = self._call_reward(gp, gr)
        contains_eos = torch.tensor(
            [cfg.stop_token_id is not None and (cfg.stop_token_id in r) for r in rows_r])
        scores = raw_scores.clone()
        if cfg.missing_eos_penalty is not None and cfg.stop_token_id is not None:
            scores = torch.where(contains_eos, scores, scores - cfg.missing_eos_penalty)
        ro = Rollout(prompts=rows_p, responses=rows_r, scores=scores,
                     raw_scores=raw_scores, contains_eos=contains_eos,
                     sample_n=cfg.sample_n, logprobs=rows_lp)
        if (getattr(cfg, "dp_rebalance_rollout", False) and self.world > 1
                and not self.algo.greedy_baseline):
            # (ReMax's greedy baseline is positional per prompt — rebalance
            # would desync it, so it stays local for that algorithm)
            ro = self._rebalance_rollout(ro)
        return ro, greedy_scores

    def _rebalance_rollout(self, ro: Rollout) -> Rollout:
        """Re-assign whole sample-groups across DP ranks so per-rank token
        totals are near-equal (greedy LPT, deterministic on every rank).
        Groups stay intact — group-relative advantages (GRPO/RLOO) and
        keep-1-of-N operate within a group, so relocation is transparent.
        Rewards/penalties were already applied locally and travel with the
        rows.  ReMax greedy scores are per-row too but its spec consumes
        them positionally, so rebalance is limited to algos without a
        greedy baseline (guarded by the caller's flag; documented)."""
        import torch.distributed as _dist
        n = max(ro.sample_n, 1)
        n_groups = ro.num_rows // n
        groups = []
        for g in range(n_groups):
            rows = list(range(g * n, (g + 1) * n))
            groups.append({
                "prompts": [ro.prompts[i] for i in rows],
                "responses": [ro.responses[i] for i in rows],
                "scores": [float(ro.scores[i]) for i in rows],
                "raw": [float(ro.raw_scores[i]) for i in rows],
                "eos": [bool(ro.contains_eos[i]) for i in rows],
                "lp": ([ro.logprobs[i] for i in rows] if ro.logprobs else None),
            })
        gathered = [None] * self.world
        _dist.all_gather_object(gathered, groups)
        flat = []
        for src, gl in enumerate(gathered):
            for gi, g in enumerate(gl):
                toks = sum(len(p) + len(r)
                           for p, r in zip(g["prompts"], g["responses"]))
                flat.append((toks, src, gi, g))
        # deterministic LPT: longest group first onto the least-loaded rank
        flat.sort(key=lambda t: (-t[0], t[1], t[2]))
        loads = [0] * self.world
        counts = [0] * self.world
        mine = []
        for toks, src, gi, g in flat:
            dst = min(range(self.world), key=lambda r: (loads[r], counts[r], r))
            loads[dst] += toks
            counts[dst] += 1
            if dst == self.rank:
                mine.append(g)
        prompts, responses, scores, raw, eos, lps = [], [], [], [], [], []
        for g in mine:
            prompts.extend(g["prompts"])
            responses.extend(g["responses"])
            scores.extend(g["scores"])
            raw.extend(g["raw"])
            eos.extend(g["eos"])
            if g["lp"] is not None:
                lps.extend(g["lp"])
        return Rollout(prompts=prompts, responses=responses,
                       scores=torch.tensor(scores, dtype=torch.float32),
                       raw_scores=torch.tensor(raw, dtype=torch.float32),
                       contains_eos=torch.tensor(eos, dtype=torch.bool),
                       sample_n=ro.sample_n,
                       logprobs=lps if lps else None)

    def _call_reward(self, rows_p: list[list[int]], rows_r: list[list[int]]) -> torch.Tensor:
        """Dispatch to the reward plug-in.  String-contract rewards
        (rewards.StringReward) get the response ids too in r1 mode, matching
        reference grpo_r1.py:250's (strings, responses_ids, tokenizer)."""
        seqs = [p + r for p, r in zip(rows_p, rows_r)]
        if getattr(self.reward_fn, "mode", None) == "r1":
            return self.reward_fn(seqs, rows_r).float()
        return self.reward_fn(seqs).float()

    # --------------------------------------------------------------- scoring
    def _pack(self, prompts, responses):
        """Pack rows into the varlen layout + response-token index maps.

        Vectorized (torch ops on CPU, single H2D copies): the original
        per-token Python list build cost ~1-2 s per full-scale update."""
        device = self.device
        R = len(prompts)
        pl = torch.tensor([len(p) for p in prompts], dtype=torch.long)
        rl = torch.tensor([len(r) for r in responses], dtype=torch.long)
        lens = pl + rl
        cu_cpu = torch.zeros(R + 1, dtype=torch.long)
        cu_cpu[1:] = torch.cumsum(lens, 0)
        T = int(cu_cpu[-1])
        ids_cpu = torch.empty(T, dtype=torch.long)
        for i, (p, r) in enumerate(zip(prompts, responses)):
            s0 = int(cu_cpu[i])
            ids_cpu[s0: s0 + len(p)] = torch.as_tensor(p, dtype=torch.long)
            ids_cpu[s0 + len(p): s0 + len(p) + len(r)] = torch.as_tensor(r, dtype=torch.long)
        # positions 0..len-1 per row
        pos_cpu = torch.arange(T) - torch.repeat_interleave(cu_cpu[:-1], lens)
        # response-token maps: hidden index (pl-1+t within row), label, row, col
        row_of_cpu = torch.repeat_interleave(torch.arange(R), rl)
        rcu = torch.zeros(R + 1, dtype=torch.long)
        rcu[1:] = torch.cumsum(rl, 0)
        TR = int(rcu[-1])
        col_of_cpu = torch.arange(TR) - torch.repeat_interleave(rcu[:-1], rl)
        flat_idx_cpu = (cu_cpu[:-1] + pl - 1)[row_of_cpu] + col_of_cpu
        flat_labels_cpu = ids_cpu[(cu_cpu[:-1] + pl)[row_of_cpu] + col_of_cpu]
        return (ids_cpu.to(device), cu_cpu.to(torch.int32).to(device),
                int(lens.max()) if R else 0, pos_cpu.to(device),
                flat_idx_cpu.to(device), flat_labels_cpu.to(device),
                row_of_cpu.to(device), col_of_cpu.to(device))

    def rollout_lp_for(self, ro: Rollout, rows: list[int]):
        """Per-row sampler logprobs when cfg.use_rollout_logprobs, else None."""
        if not getattr(self.cfg, "use_rollout_logprobs", False) or ro.logprobs is None:
            return None
        return [ro.logprobs[i] for i in rows]

    @torch.no_grad()
    def score_rows(self, prompts, responses, with_ref: bool = True,
                   with_values: bool = False, rollout_lp=None):
        """Policy (+ref) logprobs and entropy for response tokens, padded
        [R, Lmax]; the reference's chunked scoring pass (grpo_trainer.py:534-577)
        with token-budget buckets (r1's _create_batches) instead of fixed rows."""
        cfg = self.cfg
        R = len(prompts)
        Lmax = max((len(r) for r in responses), default=1)
        device = self.device
        logprobs = torch.full((R, Lmax), F.INVALID_LOGPROB, device=device)
        ref_logprobs = torch.full((R, Lmax), F.INVALID_LOGPROB, device=device) if with_ref else None
        entropy = torch.zeros((R, Lmax), device=device)
        values = torch.zeros((R, Lmax), device=device) if with_values else None
        mask = torch.zeros((R, Lmax), device=device)
        lens = [len(p) + len(r) for p, r in zip(prompts, responses)]
        if self.ref_policy is not None and with_ref:
            self.offload.model_to_device(self.ref_policy)
            self.offload.join_compute()
        if rollout_lp is not None:
            # behavior-policy logprobs came from the sampler itself
            for i, lps in enumerate(rollout_lp):
                n = len(responses[i])
                if n:
                    logprobs[i, :n] = torch.tensor(lps[:n], device=device)
                    mask[i, :n] = 1.0
        for bucket in create_batches(lens, cfg.score_token_budget):
            bp = [prompts[i] for i in bucket]
            br = [responses[i] for i in bucket]
            ids, cu, mx, pos, fidx, flab, frow, fcol = self._pack(bp, br)
            rows = torch.tensor(bucket, device=device)[frow]
            if rollout_lp is None:
                ctx = AttnContext(mode="train", positions=pos, cu_seqlens=cu, max_seqlen=mx)
                hidden = self.policy(ids, ctx)
                lp, ent = ops.token_logprob_entropy(hidden[fidx], self.policy.lm_head_weight,
                                                    flab, cfg.temperature)
                logprobs[rows, fcol] = lp.to(logprobs.dtype)
                entropy[rows, fcol] = ent.to(entropy.dtype)
            mask[rows, fcol] = 1.0
            if with_ref and self.ref_policy is not None:
                rctx = AttnContext(mode="train", positions=pos, cu_seqlens=cu, max_seqlen=mx)
                rhidden = self.ref_policy(ids, rctx)
                rlp, _ = ops.token_logprob_entropy(rhidden[fidx], self.ref_policy.lm_head_weight,
                                                   flab, cfg.temperature)
                ref_logprobs[rows, fcol] = rlp.to(ref_logprobs.dtype)
            if with_values and self.value_model is not None:
                vctx = AttnContext(mode="train", positions=pos, cu_seqlens=cu, max_seqlen=mx)
                v = self.value_model(ids, vctx)[:, 0]
                values[rows, fcol] = v[fidx].float()
        if self.ref_policy is not None and with_ref and self.offload.should_offload():
            self.offload.model_to_host(self.ref_policy)
        return logprobs, ref_logprobs, entropy, mask, values

    # ---------------------------------------------------------------- update
    def _update(self, td: TrainData) -> dict:
        cfg = self.cfg
        R = len(td.rows)
        stats = {k: [] for k in ("pg_loss", "vf_loss", "approxkl", "clipfrac",
                                 "vf_clipfrac", "entropy", "ratio", "ratio_var")}
        mb_rows_target = self.sizes["local_mini_batch_size"]
        lens = [len(p) + len(r) for p, r in zip(td.prompts, td.responses)]
        # DP ranks MUST run the same number of minibatches: each minibatch
        # boundary fires one bucketed all-reduce per parameter, so a rank
        # with fewer rows (sparse-GRPO filtering is data-dependent,
        # grpo_r1_trainer.py:565-568) would otherwise hang the others.
        n_mb_local = max(1, (R + mb_rows_target - 1) // mb_rows_target)
        n_mb = n_mb_local
        if self.world > 1:
            import torch.distributed as _dist
            t = torch.tensor([n_mb_local], dtype=torch.long,
                             device=pdist.metric_device())
            _dist.all_reduce(t, op=_dist.ReduceOp.MAX)
            n_mb = int(t.item())
        for epoch in range(cfg.num_ppo_epochs):
            perm = torch.randperm(R, generator=self._keep_gen).tolist()
            for mb_i in range(n_mb):
                mb_start = mb_i * mb_rows_target
                mb_rows = perm[mb_start: mb_start + mb_rows_target]
                zero_weight = False
                if not mb_rows:
                    # surplus minibatch on this rank: run one row at zero
                    # loss weight so gradient collectives stay matched
                    mb_rows = [perm[0] if perm else 0]
                    zero_weight = True
                budget = cfg.train_token_budget
                if budget > 0:
                    micro_batches = [[mb_rows[i] for i in b] for b in create_batches(
                        [lens[i] for i in mb_rows], budget)]
                else:
                    k = cfg.per_device_train_batch_size
                    micro_batches = [mb_rows[i:i + k] for i in range(0, len(mb_rows), k)]
                self.optimizer.zero_grad(set_to_none=True)
                for mi, micro in enumerate(micro_batches):
                    is_last = mi == len(micro_batches) - 1
                    sync_ctx = self.reducer.no_sync() if not is_last else _nullctx()
                    with sync_ctx:
                        loss, st = self._micro_step(td, micro, len(mb_rows),
                                                    zero_weight=zero_weight)
                    for k2, v in st.items():
                        if k2 in stats:
                            stats[k2].append(v)
                    if cfg.step_every_microbatch and not is_last:
                        # reference quirk mode (grpo_trainer.py:692)
                        self.optimizer.step()
                        self.optimizer.zero_grad(set_to_none=True)
                self.reducer.finalize()
                if cfg.max_grad_norm:
                    torch.nn.utils.clip_grad_norm_(
                        [p for g in self.optimizer.param_groups for p in g["params"]],
                        cfg.max_grad_norm)
                lr_now = self._lr()
                for g in self.optimizer.param_groups:
                    g["lr"] = lr_now * g["initial_lr"] / cfg.learning_rate
                self.optimizer.step()
                self.optimizer.zero_grad(set_to_none=True)
                self.lr_step += 1
        out = {}
        for k, v in stats.items():
            if v:
                out[k] = sum(v) / len(v)
        return out

    def _micro_step(self, td: TrainData, micro: list[int], mb_size: int,
                    zero_weight: bool = False):
        cfg = self.cfg
        mp = [td.prompts[i] for i in micro]
        mr = [td.responses[i] for i in micro]
        ids, cu, mx, pos, fidx, flab, frow, fcol = self._pack(mp, mr)
        sel = torch.tensor(micro, dtype=torch.long, device=self.device)
        mb = {
            "old_logprobs": td.old_logprobs[sel],
            "ref_logprobs": td.ref_logprobs[sel] if td.ref_logprobs is not None else None,
            "mask": td.mask[sel],
            "advantages": td.advantages[sel],
            "values": td.values[sel] if td.values is not None else None,
            "returns": td.returns[sel] if td.returns is not None else None,
        }
        Lmax_mb = mb["mask"].shape[1]
        ctx = AttnContext(mode="train", positions=pos, cu_seqlens=cu, max_seqlen=mx)
        hidden = self.policy(ids, ctx)
        lp_flat, ent_flat = ops.token_logprob_entropy(hidden[fidx], self.policy.lm_head_weight,
                                                      flab, cfg.temperature)
        new_logprobs = torch.full((len(micro), Lmax_mb), F.INVALID_LOGPROB,
                                  device=self.device)
        new_logprobs = new_logprobs.index_put((frow, fcol), lp_flat.to(new_logprobs.dtype))
        vpred = None
        if self.algo.needs_value and self.value_model is not None:
            vctx = AttnContext(mode="train", positions=pos, cu_seqlens=cu, max_seqlen=mx)
            v = self.value_model(ids, vctx)[:, 0]
            vpred = torch.zeros((len(micro), Lmax_mb), device=self.device)
            vpred = vpred.index_put((frow, fcol), v[fidx].float())
        loss, st = self.algo.loss(self, td, mb, new_logprobs, vpred)
        # scale: token-budget buckets re-scale by rows/minibatch rows
        # (grpo_r1_trainer.py:787-790); fixed micro-batches average over count.
        scale = 0.0 if zero_weight else len(micro) / mb_size
        (loss * scale).backward()
        with torch.no_grad():
            ratio = st.pop("ratio", None)
            stats = {"pg_loss": float(loss.detach()),
                     "approxkl": float(st.get("approxkl", 0.0)),
                     "clipfrac": float(st.get("pg_clipfrac", 0.0)),
                     "entropy": float(F.masked_mean(ent_flat, torch.ones_like(ent_flat)))
                     if ent_flat.numel() else 0.0}
            if ratio is not None:
                stats["ratio"] = float(ratio.mean())
                stats["ratio_var"] = float(ratio.var()) if ratio.numel() > 1 else 0.0
            if "vf_loss" in st:
                stats["vf_loss"] = float(st["vf_loss"])
                stats["vf_clipfrac"] = float(st.get("vf_clipfrac", 0.0))
        return loss.detach(), stats

    # ------------------------------------------------------------------ train
    def train(self, num_updates: int | None = None):
        cfg = self.cfg
        n_updates = num_updates or self.sizes["num_updates"]
        # sync a run timestamp across ranks (reference broadcast, :241-242)
        run_tag = pdist.broadcast_scalar(time.time(), device=pdist.metric_device())
        if self.eval_fn is not None and cfg.eval_at_start:
            ev = {f"initial_{k}": v for k, v in self.eval_fn(self).items()}
            self.logger.log(ev, 0)
        stop = False
        for update in range(1, n_updates + 1):
            try:
                t0 = time.time()
                if cfg.offload_optimizer:
                    # optimizer state → host for the rollout+scoring phases
                    # (reference state_to_device, grpo_trainer.py:475)
                    self.offload.optimizer_state_to(self.optimizer, "cpu")
                ro, greedy_scores = self._rollout(update)
                with self.timers.phase("score"):
                    td = self.algo.make_train_data(self, ro, greedy_scores)
                if cfg.offload_optimizer:
                    # back to HBM before the update (reference :625); the
                    # compute stream waits on the async copies, host doesn't
                    self.offload.optimizer_state_to(self.optimizer, self.device)
                    self.offload.join_compute()
                with self.timers.phase("update"):
                    upd_stats = self._update(td)
            except Exception:
                # crash recovery: persist what we have before surfacing the
                # error (the reference loses the run past the last save —
                # SURVEY §5 "Failure detection: none")
                if self.global_step > 0:
                    self.save()
                raise
            self.global_step += 1
            self.episode += self.sizes["batch_size"]
            dt = time.time() - t0
            self._log_update(ro, td, upd_stats, dt)
            if self.cfg.lr_scheduler_type == "reduce_lr_on_plateau":
                v = self._last_metrics.get("objective/rlhf_reward_old", 0.0)
                if self._plateau_best is None or v > self._plateau_best:
                    self._plateau_best = v
                    self._plateau_bad = 0
                else:
                    self._plateau_bad += 1
                    if self._plateau_bad >= self.cfg.plateau_patience:
                        self._lr_scale *= self.cfg.plateau_factor
                        self._plateau_bad = 0
            for cb in self.callbacks:
                if cb.on_update_end(self, self._last_metrics):
                    stop = True
            if (self.eval_fn is not None and cfg.eval_steps
                    and self.global_step % cfg.eval_steps == 0):
                ev = {f"eval_{k}_new": v for k, v in self.eval_fn(self).items()}
                self.logger.log(ev, self.global_step)
                self._last_metrics.update(ev)
            if cfg.save_steps and (self.global_step % cfg.save_steps == 0):
                ck = self.save()
                for cb in self.callbacks:
                    cb.on_save(self, ck)
            if stop:
                break
        return self

    # ---------------------------------------------------------------- logging
    def _log_update(self, ro: Rollout, td: TrainData, upd: dict, dt: float):
        cfg = self.cfg
        dev = pdist.metric_device()
        resp_lens = [len(r) for r in ro.responses]
        m = {
            # reference metric names (grpo_trainer.py:727-747)
            "objective/rlhf_reward_old": pdist.gather_mean(float(ro.scores.mean()), dev),
            "eval_objective/rlhf_reward_old": pdist.gather_mean(float(ro.scores.mean()), dev),
            "objective/scores_old": pdist.gather_mean(float(ro.raw_scores.mean()), dev),
            "objective/kl_old": pdist.gather_mean(td.stats.get("kl_old", 0.0), dev),
            "objective/entropy_old": pdist.gather_mean(td.stats.get("entropy_old", 0.0), dev),
            "loss/policy_avg_new": pdist.gather_mean(upd.get("pg_loss", 0.0), dev),
            "policy/approxkl_avg_new": pdist.gather_mean(upd.get("approxkl", 0.0), dev),
            "policy/clipfrac_avg_new": pdist.gather_mean(upd.get("clipfrac", 0.0), dev),
            "policy/entropy_avg_new": pdist.gather_mean(upd.get("entropy", 0.0), dev),
            "val/ratio_new": pdist.gather_mean(upd.get("ratio", 1.0), dev),
            "val/ratio_var_new": pdist.gather_mean(upd.get("ratio_var", 0.0), dev),
            "val/num_eos_tokens_old": pdist.gather_sum(float(ro.contains_eos.sum()), dev),
            "val/response_length": pdist.gather_mean(
                sum(resp_lens) / max(1, len(resp_lens)), dev),
            "lr": self._lr(),
            "episode": self.episode,
            "time/s_per_episode": dt / self.sizes["local_batch_size"],
        }
        if "vf_loss" in upd:
            m["loss/value_avg_new"] = pdist.gather_mean(upd["vf_loss"], dev)
            m["val/clipfrac_avg_new"] = pdist.gather_mean(upd.get("vf_clipfrac", 0.0), dev)
        m.update(self.timers.snapshot_and_reset())
        self.logger.log(m, self.global_step)
        if cfg.log_samples:
            rows = [{"query": ro.prompts[i][:16], "response": ro.responses[i][:24],
                     "score": float(ro.scores[i])}
                    for i in range(min(cfg.log_samples, ro.num_rows))]
            self.logger.log_samples(rows, self.global_step)
        self._last_metrics = m

    # ------------------------------------------------------------- checkpoint
    def load_checkpoint(self, ckpt_dir: str):
        """Resume policy adapter + optimizer + RNG from a checkpoint dir
        (an improvement over the reference, whose custom train() never wires
        resume_from_checkpoint — SURVEY.md §5 Checkpoint/resume)."""
        import json
        import os
        from ..utils.checkpoint import CheckpointManager, load_rng_state
        state = CheckpointManager.load_policy_state(ckpt_dir)
        missing, unexpected = self.policy.load_state_dict(state, strict=False)
        if unexpected:
            raise RuntimeError(f"unexpected keys in checkpoint: {unexpected[:5]}")
        opt_path = os.path.join(ckpt_dir, "optimizer.pt")
        if os.path.exists(opt_path):
            self.optimizer.load_state_dict(torch.load(opt_path, weights_only=False))
        if os.path.exists(os.path.join(ckpt_dir, "rng_state.pth")):
            load_rng_state(ckpt_dir)
        with open(os.path.join(ckpt_dir, "trainer_state.json")) as f:
            ts = json.load(f)
        self.global_step = ts["global_step"]
        self.episode = ts["episode"]
        vm_path = os.path.join(ckpt_dir, "value_model", "pytorch_model.bin")
        if self.value_model is not None and os.path.exists(vm_path):
            self.value_model.load_state_dict(torch.load(vm_path, weights_only=True))
        return self

    def save(self) -> str | None:
        if self.rank != 0:
            return None
        policy_state = {k: v.detach().cpu() for k, v in self.policy.state_dict().items()
                        if ("lora_" in k) or not self.cfg.use_lora}
        # modules_to_save (embed/lm_head) when LoRA is on
        if self.cfg.use_lora:
            for k, v in self.policy.state_dict().items():
                if "embed_tokens" in k or "lm_head" in k:
                    policy_state[k] = v.detach().cpu()
        value_state = None
        if self.value_model is not None:
            value_state = {k: v.detach().cpu() for k, v in self.value_model.state_dict().items()}
        return self.ckpt.save(self.global_step, self.episode, policy_state,
                              self.policy.cfg.to_dict(), self.optimizer, None,
                              value_state, getattr(self, "_last_metrics", None))


def _decay_split(model: torch.nn.Module):
    """HF get_decay_parameter_names semantics (PPO/ppo_trainer.py:351-352):
    weight decay applies to every trainable param EXCEPT biases and
    normalization weights."""
    decay, nodecay = [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        leaf = name.rsplit(".", 1)[-1]
        is_norm = "norm" in name.rsplit(".", 2)[-2] if "." in name else False
        if leaf == "bias" or is_norm:
            nodecay.append(p)
        else:
            decay.append(p)
    return decay, nodecay


class _nullctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
