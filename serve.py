"""Minimal serving front end over the in-process sampler engine.

The reference has no serving story (its vLLM engine is booted per training
update and thrown away); this exposes the same MI355X-native engine that
drives training rollouts — paged fp8/bf16 KV, continuous batching,
hipGraph-captured decode — behind a small HTTP API:

    python serve.py --model /path/to/hf-qwen2-checkpoint [--port 8000]
    python serve.py --preset qwen2.5-1.5b            # random-init (smoke)

Endpoints:
    POST /generate   {"prompt": str | [str], "max_tokens": int,
                      "temperature": float, "top_p": float, "n": int}
    GET  /health

Requests are micro-batched: the server collects whatever is queued, runs
one engine.generate over the batch, and answers everyone (the engine's
continuous batching handles ragged lengths internally).
"""
from __future__ import annotations

import argparse
import asyncio
import os
import sys
import threading

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from nanorlhf_amd.models import CausalLM, get_config
from nanorlhf_amd.sampler import SamplerEngine, SamplingParams


class Server:
    def __init__(self, model, tokenizer, kv_pool_tokens: int,
                 kv_cache_dtype: str = "fp8_e4m3"):
        self.model = model.eval()
        self.tokenizer = tokenizer
        dev = next(model.parameters()).device
        self.engine = SamplerEngine(
            model, kv_pool_tokens=kv_pool_tokens,
            kv_cache_dtype=kv_cache_dtype if dev.type == "cuda" else "bf16")
        self.lock = threading.Lock()   # one engine pass at a time

    def encode(self, prompt: str) -> list[int]:
        if self.tokenizer is not None:
            return self.tokenizer(prompt)["input_ids"]
        # tokenizer-less smoke mode: bytes as ids (mod vocab)
        V = self.model.cfg.vocab_size
        return [2 + (b % (V - 3)) for b in prompt.encode()][:512]

    def decode(self, ids: list[int]) -> str:
        if self.tokenizer is not None:
            return self.tokenizer.decode(ids, skip_special_tokens=True)
        return " ".join(str(t) for t in ids)

    def generate(self, prompts: list[str], max_tokens: int, temperature: float,
                 top_p: float, n: int) -> list[list[str]]:
        enc = [self.encode(p) for p in prompts]
        params = SamplingParams(
            n=n, temperature=temperature, top_p=top_p, max_tokens=max_tokens,
            seed=int.from_bytes(os.urandom(4), "little"),
            stop_token_id=(self.tokenizer.eos_token_id
                           if self.tokenizer is not None else None))
        with self.lock, torch.no_grad():
            out = self.engine.generate(enc, params,
                                       pad_token_id=0, merge_lora=True)
        pad = 0
        results: list[list[str]] = []
        for pi in range(len(prompts)):
            outs = []
            for j in range(n):
                row = out[pi * n + j].tolist()
                stop = params.stop_token_id
                if stop is not None and stop in row:
                    row = row[: row.index(stop)]
                else:
                    while row and row[-1] == pad:
                        row.pop()
                outs.append(self.decode(row))
            results.append(outs)
        return results


def build_app(server: Server):
    from fastapi import Body, FastAPI

    app = FastAPI(title="nanorlhf_amd serve")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(next(server.model.parameters()).device)}

    @app.post("/generate")
    async def generate(req: dict = Body(...)):
        prompt = req.get("prompt", "")
        prompts = [prompt] if isinstance(prompt, str) else list(prompt)
        loop = asyncio.get_event_loop()
        results = await loop.run_in_executor(
            None, server.generate, prompts,
            int(req.get("max_tokens", 128)), float(req.get("temperature", 0.7)),
            float(req.get("top_p", 0.95)), int(req.get("n", 1)))
        return {"completions": results}

    return app


def make_server(model_path: str | None, preset: str, kv_pool_tokens: int,
                kv_cache_dtype: str = "fp8_e4m3") -> Server:
    tokenizer = None
    if model_path:
        from nanorlhf_amd.data.tokenizer import load_tokenizer
        from nanorlhf_amd.models.hf_import import load_pretrained
        model = load_pretrained(model_path)
        tokenizer = load_tokenizer(model_path)
    else:
        cfg = get_config(preset, **({} if torch.cuda.is_available()
                                    else {"dtype": "float32"}))
        model = CausalLM(cfg)
    if torch.cuda.is_available():
        model = model.to("cuda").to(torch.bfloat16)
    return Server(model, tokenizer, kv_pool_tokens, kv_cache_dtype)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", type=str, default=None,
                    help="local HF Qwen2 checkpoint dir (with tokenizer files)")
    ap.add_argument("--preset", type=str, default="qwen2.5-1.5b")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--host", type=str, default="127.0.0.1")
    ap.add_argument("--kv-pool-tokens", type=int, default=2_000_000)
    ap.add_argument("--kv-dtype", type=str, default="fp8_e4m3")
    args = ap.parse_args()
    import uvicorn
    server = make_server(args.model, args.preset, args.kv_pool_tokens,
                         args.kv_dtype)
    uvicorn.run(build_app(server), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
