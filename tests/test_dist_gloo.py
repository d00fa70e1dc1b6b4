"""Multi-process distributed tests over gloo (world_size=2, CPU).

Covers the DP seam the driver exercises at 1/2/4/8 GPUs with RCCL: the
GradReducer's bucketed overlap must equal a plain all-reduce, and gathered
metrics must average across ranks."""
import os

import pytest
import torch
import torch.distributed as torch_dist
import torch.multiprocessing as mp


def _init_pg(rank, world, store_path):
    """File-store rendezvous: no TCP port collisions across repeated runs."""
    torch_dist.init_process_group("gloo", rank=rank, world_size=world,
                                  init_method=f"file://{store_path}")


def _run_reducer(rank, world, store_path, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    _init_pg(rank, world, store_path)
    try:
        from nanorlhf_amd.parallel.ddp import GradReducer
        from nanorlhf_amd.parallel import dist as pdist

        torch.manual_seed(100 + rank)
        model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
        # identical weights via broadcast
        for p in model.parameters():
            torch_dist.broadcast(p.data, 0)
        reducer = GradReducer(model.parameters(), bucket_bytes=128)
        x = torch.randn(4, 8)
        # accumulate 2 micro-batches; only last syncs
        with reducer.no_sync():
            model(x).sum().backward()
        model(x * 2).sum().backward()
        reducer.finalize()
        # oracle: replay with explicit all-reduce
        model2 = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
        model2.load_state_dict(model.state_dict())
        model2.zero_grad()
        model2(x).sum().backward()
        model2(x * 2).sum().backward()
        for p in model2.parameters():
            torch_dist.all_reduce(p.grad)
            p.grad /= world
        ok = all(torch.allclose(p1.grad, p2.grad, atol=1e-6)
                 for p1, p2 in zip(model.parameters(), model2.parameters()))
        gm = pdist.gather_mean(float(rank))
        ok = ok and abs(gm - 0.5) < 1e-9
        q.put((rank, ok))
    finally:
        torch_dist.destroy_process_group()


def _run_trainer_dp(rank, world, store_path, q, tmpdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    _init_pg(rank, world, store_path)
    try:
        import torch
        from nanorlhf_amd.algos import reinforce
        from nanorlhf_amd.algos.reinforce import ReinforceConfig
        from nanorlhf_amd.data import hh_shaped_prompts
        from nanorlhf_amd.models import CausalLM
        from nanorlhf_amd.rewards import constant_reward

        cfg = ReinforceConfig(model_preset="tiny", dtype="float32", use_lora=True,
                              lora_r=4, lora_alpha=8, per_device_train_batch_size=2,
                              gradient_accumulation_steps=1, num_mini_batches=2,
                              total_episodes=8, response_length=4, temperature=1.0,
                              stop_token_id=1, output_dir=os.path.join(tmpdir, "dp"),
                              gradient_checkpointing=False, score_token_budget=256)
        torch.manual_seed(0)
        policy = CausalLM.from_preset("tiny")
        ref = CausalLM.from_preset("tiny")
        ref.load_state_dict(policy.state_dict())
        prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=8)
        tr = reinforce.make_trainer(cfg, policy, ref, lambda s: constant_reward(s), prompts)
        tr.train(num_updates=1)
        # trainable params must be identical across ranks after the synced update
        flat = torch.cat([p.detach().reshape(-1) for p in tr.policy.parameters()
                          if p.requires_grad])
        flats = [torch.zeros_like(flat) for _ in range(world)]
        torch_dist.all_gather(flats, flat)
        ok = all(torch.allclose(flats[0], f, atol=1e-6) for f in flats)
        q.put((rank, ok))
    finally:
        if torch_dist.is_initialized():
            torch_dist.destroy_process_group()


def _run_sparse_grpo_dp(rank, world, store_path, q, tmpdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    _init_pg(rank, world, store_path)
    try:
        import torch
        from nanorlhf_amd.algos import grpo
        from nanorlhf_amd.algos.grpo import GRPOConfig
        from nanorlhf_amd.data import hh_shaped_prompts
        from nanorlhf_amd.models import CausalLM

        cfg = GRPOConfig(model_preset="tiny", dtype="float32", use_lora=True,
                         lora_r=4, lora_alpha=8, per_device_train_batch_size=2,
                         gradient_accumulation_steps=1, num_mini_batches=2,
                         total_episodes=8, sample_n=2, response_length=4,
                         temperature=1.0, stop_token_id=1,
                         output_dir=os.path.join(tmpdir, f"sp{rank}"),
                         gradient_checkpointing=False, score_token_budget=256,
                         sparse_filter=True, train_token_budget=64)
        torch.manual_seed(0)
        policy = CausalLM.from_preset("tiny")
        ref = CausalLM.from_preset("tiny")
        ref.load_state_dict(policy.state_dict())
        prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=8)

        def reward(seqs):
            # rank 1 gets ALL-ZERO scores -> sparse filter drops everything
            # there while rank 0 keeps rows: minibatch counts must still match
            if int(os.environ["RANK"]) == 1:
                return torch.zeros(len(seqs))
            return torch.tensor([(sum(s) + len(s)) % 3 - 1.0 for s in seqs])

        tr = grpo.make_trainer(cfg, policy, ref, reward, prompts)
        tr.train(num_updates=1)
        q.put((rank, True))
    finally:
        if torch_dist.is_initialized():
            torch_dist.destroy_process_group()


def _run_rebalance_dp(rank, world, store_path, q, tmpdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    _init_pg(rank, world, store_path)
    try:
        import torch
        from nanorlhf_amd.algos import grpo
        from nanorlhf_amd.algos.grpo import GRPOConfig
        from nanorlhf_amd.algos.trainer import Rollout
        from nanorlhf_amd.data import hh_shaped_prompts
        from nanorlhf_amd.models import CausalLM

        cfg = GRPOConfig(model_preset="tiny", dtype="float32", use_lora=True,
                         lora_r=4, lora_alpha=8, per_device_train_batch_size=2,
                         gradient_accumulation_steps=1, num_mini_batches=2,
                         total_episodes=8, sample_n=2, response_length=4,
                         temperature=1.0, stop_token_id=1,
                         output_dir=os.path.join(tmpdir, f"rb{rank}"),
                         gradient_checkpointing=False, score_token_budget=256,
                         dp_rebalance_rollout=True)
        torch.manual_seed(0)
        policy = CausalLM.from_preset("tiny")
        ref = CausalLM.from_preset("tiny")
        ref.load_state_dict(policy.state_dict())
        prompts = hh_shaped_prompts(16, 1024, min_len=4, max_len=8)
        tr = grpo.make_trainer(cfg, policy, ref,
                               lambda s: torch.tensor([float(len(x) % 3) for x in s]),
                               prompts)

        # unit check of the rebalancer: rank 0 holds LONG groups, rank 1
        # short — after rebalance token loads must be near-equal and the
        # global multiset of groups preserved
        L = 40 if rank == 0 else 4
        n_groups = 4
        ps, rs = [], []
        for g in range(n_groups):
            for j in range(2):
                ps.append([rank * 1000 + g * 10 + j] * 3)
                rs.append([5] * L)
        ro = Rollout(prompts=ps, responses=rs,
                     scores=torch.arange(len(ps), dtype=torch.float32),
                     raw_scores=torch.arange(len(ps), dtype=torch.float32),
                     contains_eos=torch.zeros(len(ps), dtype=torch.bool),
                     sample_n=2,
                     logprobs=[[0.1] * L for _ in ps])
        ro2 = tr._rebalance_rollout(ro)
        my_toks = sum(len(p) + len(r) for p, r in zip(ro2.prompts, ro2.responses))
        toks = [torch.zeros(1) for _ in range(world)]
        torch_dist.all_gather(toks, torch.tensor([float(my_toks)]))
        total = sum(float(t) for t in toks)
        ok = all(abs(float(t) - total / world) <= 90 for t in toks)
        # global group preservation: gather all first-prompt markers
        markers = sorted(p[0] for p in ro2.prompts)
        allm = [None] * world
        torch_dist.all_gather_object(allm, markers)
        merged = sorted(x for m in allm for x in m)
        want = sorted([r * 1000 + g * 10 + j for r in range(world)
                       for g in range(4) for j in range(2)])
        ok = ok and merged == want
        # and a full update still runs with the flag on
        tr.train(num_updates=1)
        q.put((rank, ok))
    finally:
        if torch_dist.is_initialized():
            torch_dist.destroy_process_group()


@pytest.mark.parametrize("fn", [_run_reducer, _run_trainer_dp, _run_sparse_grpo_dp, _run_rebalance_dp])
def test_world2_gloo(fn, tmp_path):
    world = 2
    store_path = str(tmp_path / "pg_store")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    args = (world, store_path, q) if fn is _run_reducer else \
        (world, store_path, q, str(tmp_path))
    procs = [ctx.Process(target=fn, args=(r, *args)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, ok = q.get(timeout=300)
        results[rank] = ok
    for p in procs:
        p.join(timeout=60)
    assert all(results.values()), results
