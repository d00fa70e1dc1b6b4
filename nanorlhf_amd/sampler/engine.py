"""In-process generation engine — the vLLM replacement.

Deletes the reference's per-update disk round trip (save LoRA → reload base
→ merge_and_unload → save merged → boot LLM() → generate → del llm,
GRPO/grpo_trainer.py:122-166): here the sampler runs the live policy module
in-place (LoRA pre-merged into an HBM buffer, models/lora.py), with a paged
KV pool, continuous batching (finished sequences free pages, queued ones are
admitted), prefill via the varlen MFMA flash kernel and decode via the paged
LDS-staged attention kernel, temperature/top-p sampling in HIP.

API mirrors the reference's `vllm_generate(N, model, tokenizer, prompts,
temperature, max_tokens)` contract (grpo_trainer.py:122-166): `generate`
returns per-prompt responses padded to max_tokens with the pad token.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from .. import ops
from ..models.qwen2 import AttnContext, CausalLM
from ..models.lora import merge_for_rollout, unmerge
from .cache import PagedKVCache, SeqState
from .decode_batch import DecodeBatch


@dataclass
class SamplingParams:
    n: int = 1
    temperature: float = 1.0
    top_p: float = 0.95
    max_tokens: int = 128
    seed: int = 0
    stop_token_id: int | None = None  # EOS; None → run to max_tokens


class SamplerEngine:
    KV_DTYPES = {"bf16": torch.bfloat16, "fp8_e4m3": torch.float8_e4m3fn}

    def __init__(self, model: CausalLM, kv_pool_tokens: int, page_size: int = 16,
                 max_num_seqs: int = 4096, prefill_chunk_tokens: int = 131072,
                 compact_interval: int = 32, use_graphs: bool = True,
                 kv_cache_dtype: str = "bf16",
                 rollout_weight_dtype: str = "bf16"):
        self.model = model
        self.device = next(model.parameters()).device
        self.dtype = next(model.parameters()).dtype
        kv_dtype = (self.KV_DTYPES[kv_cache_dtype]
                    if self.device.type == "cuda" or kv_cache_dtype != "bf16"
                    else self.dtype)
        if self.device.type != "cuda" and kv_cache_dtype == "bf16":
            kv_dtype = self.dtype  # CPU tests may run fp32 models
        self.pool = PagedKVCache.for_budget(model.cfg, kv_pool_tokens, page_size,
                                            device=self.device, dtype=kv_dtype)
        self.max_num_seqs = max_num_seqs
        self.prefill_chunk_tokens = prefill_chunk_tokens
        self.compact_interval = compact_interval
        self.use_graphs = use_graphs and self.device.type == "cuda"
        self._graph = None
        self._graph_version = None
        self._step_dev = (torch.zeros(1, dtype=torch.long, device=self.device)
                          if self.device.type == "cuda" else None)
        # "fp8_e4m3": rollout-only e4m3 layer weights via torch._scaled_mm
        # (inference GEMMs at ~2x the bf16 rate; lm_head stays bf16 so the
        # sampled logits/logprobs come from the exact head).  Training
        # weights are never touched; validated like the fp8 KV mode.
        self.rollout_weight_dtype = (rollout_weight_dtype
                                     if self.device.type == "cuda" else "bf16")

    # ----------------------------------------------------------------- utils
    def _slots_for_range(self, seq: SeqState, start: int, end: int) -> list[int]:
        ps = self.pool.page_size
        return [seq.slot_of(p, ps) for p in range(start, end)]

    def _sample_from_hidden(self, hidden_last: torch.Tensor,
                            params: SamplingParams):
        """Returns (tokens [B], logprobs [B] fp32) — the chosen token's
        logprob under the temperature-scaled softmax (vLLM logprobs parity;
        matches the scoring pass's logits/temperature quirk)."""
        logits = self.model.logits(hidden_last)
        if self._step_dev is not None:
            # device-side step counter: correct under hipGraph replay
            self._step_dev.add_(1)
            tok, lp = ops.ext().sample_topp_dev(logits.contiguous(),
                                                float(params.temperature),
                                                float(params.top_p),
                                                int(params.seed), self._step_dev)
            return tok, lp
        self._sample_step += 1
        tok = ops.sample_tokens(logits, params.temperature, params.top_p,
                                params.seed, self._sample_step)
        inv_t = 1.0 / params.temperature if params.temperature > 0 else 1.0
        logp = torch.log_softmax(logits.float() * inv_t, dim=-1)
        lp = logp.gather(1, tok.unsqueeze(1)).squeeze(1)
        return tok, lp

    # -------------------------------------------------------------- prefill
    @torch.no_grad()
    def _prefill(self, seqs: list[SeqState], params: SamplingParams):
        """Prefill `seqs` (packed varlen, chunked by token budget), append KV,
        sample each sequence's first generated token."""
        i = 0
        while i < len(seqs):
            # pack a chunk of sequences up to the token budget
            chunk: list[SeqState] = []
            total = 0
            while i < len(seqs) and (not chunk or total + len(seqs[i]) <= self.prefill_chunk_tokens):
                total += len(seqs[i])
                chunk.append(seqs[i])
                i += 1
            lens = [len(s) for s in chunk]
            ids = torch.tensor([t for s in chunk for t in s.tokens], dtype=torch.long,
                               device=self.device)
            cu = torch.zeros(len(chunk) + 1, dtype=torch.int32, device=self.device)
            cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=self.device), 0)
            pos = torch.cat([torch.arange(n, device=self.device) for n in lens])
            slots = torch.tensor([sl for s in chunk for sl in self._slots_for_range(s, 0, len(s))],
                                 dtype=torch.long, device=self.device)
            ctx = AttnContext(mode="prefill", positions=pos, cu_seqlens=cu,
                              max_seqlen=max(lens), kv_caches=self.pool.layers, slots=slots)
            hidden = self.model(ids, ctx)
            last_idx = cu[1:].long() - 1
            tokens, lps = self._sample_from_hidden(hidden[last_idx], params)
            tok_list = tokens.tolist()
            lp_list = lps.tolist()
            for s, t, l in zip(chunk, tok_list, lp_list):
                s.tokens.append(int(t))
                s.logprobs.append(float(l))
                if params.stop_token_id is not None and int(t) == params.stop_token_id:
                    s.finished = True

    # --------------------------------------------------------------- decode
    def _decode_step_eager(self, db: DecodeBatch, params: SamplingParams):
        ids, pos, slots, seq_lens, bt = db.step_inputs()
        ctx = AttnContext(mode="decode", positions=pos,
                          kv_caches=self.pool.layers, slots=slots,
                          block_tables=bt, seq_lens=seq_lens)
        hidden = self.model(ids, ctx)
        tokens, lps = self._sample_from_hidden(hidden, params)
        db.commit(tokens, lps)

    def _run_decode_step(self, db: DecodeBatch, params: SamplingParams) -> int:
        """One decode step; hipGraph-captured and replayed when the batch
        state is stable.  Returns the number of steps actually taken (capture
        does 2 warmup steps)."""
        if not self.use_graphs:
            self._decode_step_eager(db, params)
            return 1
        if self._graph is None or self._graph_version != db.version:
            torch.cuda.synchronize()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(2):
                    self._decode_step_eager(db, params)
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._decode_step_eager(db, params)
            self._graph = g
            self._graph_version = db.version
            return 2  # the warmup steps ran; captured step executes on replay
        self._graph.replay()
        return 1

    # -------------------------------------------------------------- generate
    @torch.no_grad()
    def generate(self, prompts: list[list[int]], params: SamplingParams,
                 pad_token_id: int = 0, merge_lora: bool = True,
                 return_logprobs: bool = False):
        """Sample params.n continuations per prompt.

        Returns LongTensor [len(prompts)*n, max_tokens]: responses
        right-padded with pad_token_id, matching the reference's
        vllm_generate output contract (grpo_trainer.py:152-164).
        Row order: prompt-major (prompt0 sample0..n-1, prompt1 ...).
        """
        was_training = self.model.training
        self.model.eval()
        if merge_lora:
            merge_for_rollout(self.model,
                              quant=(self.rollout_weight_dtype
                                     if self.rollout_weight_dtype != "bf16" else None))
        try:
            waiting: list[SeqState] = []
            uid = 0
            for pi, p in enumerate(prompts):
                for j in range(params.n):
                    waiting.append(SeqState(uid, p, out_index=pi * params.n + j))
                    uid += 1
            done: list[SeqState] = []
            self._sample_step = 0
            if self._step_dev is not None:
                self._step_dev.zero_()
            self._graph = None  # new rollout -> new capture
            db = DecodeBatch(self.pool, [], params.max_tokens,
                             params.stop_token_id, self.device)
            outer_guard = 0

            def admit():
                """Move waiting → active, allocating each sequence's full page
                budget (prompt + max_tokens) up front: no oversubscription, so
                decode can never dead-lock on pages mid-flight."""
                admitted = []
                while waiting and len(db) + len(admitted) < self.max_num_seqs:
                    s = waiting[-1]
                    ps = self.pool.page_size
                    need = (len(s) + params.max_tokens + ps - 1) // ps
                    if need > self.pool.num_pages:
                        raise RuntimeError(
                            f"sequence needs {need} pages (prompt {len(s)} + "
                            f"max_tokens {params.max_tokens}) but the KV pool "
                            f"has only {self.pool.num_pages}; raise "
                            "kv_pool_tokens or lower response_length")
                    if need > self.pool.free_pages:
                        break
                    waiting.pop()
                    s.ensure_capacity(self.pool, len(s) + params.max_tokens)
                    admitted.append(s)
                return admitted

            while waiting or len(db):
                fresh = admit()
                if fresh:
                    self._prefill(fresh, params)
                    db.extend(fresh)
                # device-resident decode: no host sync inside the chunk
                inner = 0
                while len(db) and inner < self.compact_interval:
                    inner += self._run_decode_step(db, params)
                done.extend(db.compact())
                outer_guard += 1
                if outer_guard > 8 * (params.max_tokens // self.compact_interval + 2) \
                        + len(prompts) * params.n:
                    raise RuntimeError("sampler scheduling did not converge")

            out = torch.full((len(prompts) * params.n, params.max_tokens), pad_token_id,
                             dtype=torch.long)
            lp_out = torch.zeros(len(prompts) * params.n, params.max_tokens)
            for s in done:
                resp = s.response[: params.max_tokens]
                if resp:
                    out[s.out_index, : len(resp)] = torch.tensor(resp, dtype=torch.long)
                    lps = s.logprobs[: len(resp)]
                    lp_out[s.out_index, : len(lps)] = torch.tensor(lps)
            if return_logprobs:
                return out, lp_out
            return out
        finally:
            if merge_lora:
                unmerge(self.model)
            if was_training:
                self.model.train()
