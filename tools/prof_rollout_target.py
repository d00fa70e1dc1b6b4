"""Short rollout-only profile target: 256 seqs x 128 new tokens."""
import torch
from nanorlhf_amd.models import CausalLM, get_config
from nanorlhf_amd.sampler import SamplerEngine, SamplingParams
from nanorlhf_amd.data import hh_shaped_prompts

torch.manual_seed(0)
cfg = get_config("qwen2.5-1.5b")
m = CausalLM(cfg).to("cuda").to(torch.bfloat16).eval()
eng = SamplerEngine(m, kv_pool_tokens=300000, page_size=16)
prompts = hh_shaped_prompts(64, cfg.vocab_size, seed=3)
params = SamplingParams(n=4, temperature=0.7, top_p=0.95, max_tokens=128, seed=1, stop_token_id=1)
out = eng.generate(prompts, params)  # warm
torch.cuda.synchronize()
import time
t0 = time.perf_counter()
out = eng.generate(prompts, params)
torch.cuda.synchronize()
print("rollout 256seq x 128tok:", time.perf_counter() - t0, "s")
