"""Device-resident decode batch state.

The per-token decode loop keeps ALL scheduling state (sequence lengths,
block tables, slot mapping, generated-token buffer, finished mask) on the
GPU as tensors: one decode step costs a handful of small index kernels +
the model forward, with no per-sequence Python work and no host round trip
(the first bench profile showed list rebuilding at batch 2048 dominating
rollout time).  Host synchronization happens only every `compact_interval`
steps, when finished sequences are retired (pages freed) and queued
sequences admitted."""
from __future__ import annotations

import torch

from .cache import PagedKVCache, SeqState


class DecodeBatch:
    def __init__(self, pool: PagedKVCache, seqs: list[SeqState], max_tokens: int,
                 stop_token_id: int | None, device):
        self.pool = pool
        self.device = device
        self.max_tokens = max_tokens
        self.stop = stop_token_id
        self.seqs: list[SeqState] = []
        self.bt = torch.zeros(0, 1, dtype=torch.int32, device=device)
        self.seq_lens = torch.zeros(0, dtype=torch.int32, device=device)
        self.gen_count = torch.zeros(0, dtype=torch.int32, device=device)
        self.last_tok = torch.zeros(0, dtype=torch.long, device=device)
        self.finished = torch.zeros(0, dtype=torch.bool, device=device)
        self.out = torch.zeros(0, max_tokens, dtype=torch.long, device=device)
        self.lp = torch.zeros(0, max_tokens, dtype=torch.float32, device=device)
        self.version = 0  # bumped whenever state tensors are rebuilt
        if seqs:
            self.extend(seqs)

    def __len__(self):
        return len(self.seqs)

    # ------------------------------------------------------------------ build
    def _rows_for(self, seqs: list[SeqState], pad_to_pages: int):
        dev = self.device
        B = len(seqs)
        bt = torch.zeros(B, pad_to_pages, dtype=torch.int32)
        for r, s in enumerate(seqs):
            bt[r, : len(s.pages)] = torch.tensor(s.pages, dtype=torch.int32)
        seq_lens = torch.tensor([len(s) for s in seqs], dtype=torch.int32)
        last = torch.tensor([s.tokens[-1] for s in seqs], dtype=torch.long)
        fin = torch.tensor([s.finished for s in seqs], dtype=torch.bool)
        out = torch.zeros(B, self.max_tokens, dtype=torch.long)
        lp = torch.zeros(B, self.max_tokens, dtype=torch.float32)
        gen = torch.zeros(B, dtype=torch.int32)
        for r, s in enumerate(seqs):
            # prefill already produced response token(s)
            resp = s.response
            if resp:
                out[r, : len(resp)] = torch.tensor(resp, dtype=torch.long)
                lp[r, : len(s.logprobs)] = torch.tensor(s.logprobs, dtype=torch.float32)
                gen[r] = len(resp)
        return (bt.to(dev), seq_lens.to(dev), last.to(dev), fin.to(dev),
                out.to(dev), gen.to(dev), lp.to(dev))

    def extend(self, seqs: list[SeqState]):
        pages = max([len(s.pages) for s in seqs] + [self.bt.shape[1]])
        if self.bt.shape[1] < pages and len(self.seqs):
            pad = torch.zeros(len(self.seqs), pages - self.bt.shape[1],
                              dtype=torch.int32, device=self.device)
            self.bt = torch.cat([self.bt, pad], dim=1)
        bt, sl, lt, fin, out, gen, lp = self._rows_for(seqs, pages)
        self.bt = torch.cat([self.bt, bt]) if len(self.seqs) else bt
        self.seq_lens = torch.cat([self.seq_lens, sl])
        self.last_tok = torch.cat([self.last_tok, lt])
        self.finished = torch.cat([self.finished, fin])
        self.out = torch.cat([self.out, out])
        self.lp = torch.cat([self.lp, lp])
        self.gen_count = torch.cat([self.gen_count, gen])
        self.seqs.extend(seqs)
        self.version += 1

    # ------------------------------------------------------------------ step
    def step_inputs(self):
        """Tensors for the next decode forward (all on device, no sync)."""
        ps = self.pool.page_size
        pos = (self.seq_lens - 1).long()
        page_idx = torch.div(pos, ps, rounding_mode="floor")
        page = self.bt.gather(1, page_idx.unsqueeze(1).int().long()).squeeze(1).long()
        slots = page * ps + pos % ps
        return self.last_tok, pos, slots, self.seq_lens, self.bt

    def commit(self, tokens: torch.Tensor, lps: torch.Tensor | None = None):
        """Record sampled tokens; advance lengths.  Fully IN-PLACE on the
        batch's state tensors so the whole step is hipGraph-capturable (no
        host sync, no tensor reassignment)."""
        write = (~self.finished) & (self.gen_count < self.max_tokens)
        idx = self.gen_count.clamp(max=self.max_tokens - 1).long().unsqueeze(1)
        cur = self.out.gather(1, idx).squeeze(1)
        val = torch.where(write, tokens, cur)
        self.out.scatter_(1, idx, val.unsqueeze(1))
        if lps is not None:
            cur_lp = self.lp.gather(1, idx).squeeze(1)
            self.lp.scatter_(1, idx,
                             torch.where(write, lps, cur_lp).unsqueeze(1))
        wi = write.int()
        self.gen_count.add_(wi)
        self.seq_lens.add_(wi)
        torch.where(write, tokens, self.last_tok, out=self.last_tok)
        if self.stop is not None:
            self.finished.logical_or_(write & (tokens == self.stop))
        self.finished.logical_or_(self.gen_count >= self.max_tokens)

    # ------------------------------------------------------------------ sync
    def compact(self):
        """Host sync: retire finished sequences (free pages, materialize their
        responses) and drop their rows.  Returns list of retired SeqStates."""
        fin = self.finished.cpu()
        if not bool(fin.any()):
            return []
        keep_mask = ~fin
        retired_rows = fin.nonzero(as_tuple=True)[0].tolist()
        out_cpu = self.out[fin.to(self.device)].cpu()
        lp_cpu = self.lp[fin.to(self.device)].cpu()
        gen_cpu = self.gen_count[fin.to(self.device)].cpu()
        retired = []
        for j, r in enumerate(retired_rows):
            s = self.seqs[r]
            n = int(gen_cpu[j])
            resp = out_cpu[j, :n].tolist()
            if self.stop is not None and self.stop in resp:
                resp = resp[: resp.index(self.stop) + 1]
            s.tokens = s.tokens[: s.prompt_len] + resp
            s.logprobs = lp_cpu[j, : len(resp)].tolist()
            s.finished = True
            self.pool.free(s.pages)
            s.pages = []
            retired.append(s)
        km = keep_mask.to(self.device)
        self.seqs = [s for r, s in enumerate(self.seqs) if not bool(fin[r])]
        self.bt = self.bt[km]
        self.seq_lens = self.seq_lens[km]
        self.gen_count = self.gen_count[km]
        self.last_tok = self.last_tok[km]
        self.finished = self.finished[km]
        self.out = self.out[km]
        self.lp = self.lp[km]
        self.version += 1
        return retired

    def all_finished(self) -> bool:
        if not self.seqs:
            return True
        return bool(self.finished.all().item())
