"""Multi-process (gloo, world_size=2) tests of the DDP-equivalent reducer
and ZeRO-1 sharded optimizer — the distributed-correctness tier that runs
without GPUs (SURVEY.md §4 test strategy)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from tests.conftest import free_port


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _ddp_worker(rank, world, port, q):
    try:
        _init(rank, world, port)
        from relora_amd.parallel import DistributedModel

        torch.manual_seed(0)
        model = torch.nn.Sequential(
            torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4)
        )
        wrapped = DistributedModel(model, bucket_cap_mb=0.0001)  # force many buckets

        # full batch of 4; each rank takes 2 -> averaged grads must equal
        # the single-process full-batch gradient
        torch.manual_seed(42)
        x = torch.randn(4, 8)
        y = torch.randn(4, 4)
        xb = x[rank * 2 : rank * 2 + 2]
        yb = y[rank * 2 : rank * 2 + 2]

        wrapped.set_gradient_sync(True)
        loss = torch.nn.functional.mse_loss(wrapped(xb), yb)
        loss.backward()
        wrapped.finish_gradient_sync()

        # single-process reference (same init thanks to same seed + broadcast)
        ref = torch.nn.Sequential(
            torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4)
        )
        torch.manual_seed(0)
        for p_ref, p in zip(ref.parameters(), [None] * 0):
            pass
        ref.load_state_dict({k: v.clone() for k, v in model.state_dict().items()})
        ref_loss = (
            torch.nn.functional.mse_loss(ref(x[:2]), y[:2])
            + torch.nn.functional.mse_loss(ref(x[2:]), y[2:])
        ) / 2
        ref_loss.backward()

        ok = all(
            torch.allclose(p.grad, rp.grad, atol=1e-6)
            for p, rp in zip(model.parameters(), ref.parameters())
        )
        q.put((rank, ok, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ddp_grad_averaging():
    world, port = 2, free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ddp_worker, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


def _accum_worker(rank, world, port, q):
    try:
        _init(rank, world, port)
        from relora_amd.parallel import DistributedModel

        torch.manual_seed(0)
        model = torch.nn.Linear(4, 4, bias=False)
        wrapped = DistributedModel(model)
        torch.manual_seed(7)
        xs = [torch.randn(2, 4) for _ in range(4)]  # 2 micro-steps x 2 ranks

        # micro-step 1: no sync; micro-step 2: sync
        wrapped.set_gradient_sync(False)
        wrapped(xs[rank]).sum().backward()
        wrapped.set_gradient_sync(True)
        wrapped(xs[2 + rank]).sum().backward()
        wrapped.finish_gradient_sync()

        ref = torch.nn.Linear(4, 4, bias=False)
        ref.load_state_dict(model.state_dict())
        for x in xs:
            ref(x).sum().backward()
        ok = torch.allclose(model.weight.grad, ref.weight.grad / world, atol=1e-6)
        q.put((rank, ok, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ddp_grad_accumulation_boundary_only():
    world, port = 2, free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_accum_worker, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


def _zero_worker(rank, world, port, q):
    try:
        _init(rank, world, port)
        from relora_amd.parallel import DistributedModel, ZeroRedundancyAdamW

        torch.manual_seed(0)
        model = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 8))
        wrapped = DistributedModel(model)
        params = [p for p in model.parameters() if p.requires_grad]
        opt = ZeroRedundancyAdamW(params, lr=1e-2, betas=(0.9, 0.999), weight_decay=0.01)

        # reference: plain AdamW on a copy with the same (averaged) grads
        ref = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 8))
        ref.load_state_dict(model.state_dict())
        from relora_amd.ops.optim import AdamW
        ref_opt = AdamW(list(ref.parameters()), lr=1e-2, betas=(0.9, 0.999), weight_decay=0.01)

        torch.manual_seed(123)
        for it in range(3):
            x = torch.randn(4, 8)  # same on all ranks -> same grads
            wrapped.set_gradient_sync(True)
            wrapped(x).pow(2).mean().backward()
            wrapped.finish_gradient_sync()
            opt.step()

            ref(x).pow(2).mean().backward()
            ref_opt.step()

            wrapped.zero_grad_buffers()
            ref_opt.zero_grad()

        ok = all(
            torch.allclose(p, rp, atol=1e-5)
            for p, rp in zip(model.parameters(), ref.parameters())
        )
        # state_dict coverage
        opt.consolidate_state_dict()
        sd_ok = True
        if rank == 0:
            sd = opt.state_dict()
            sd_ok = len(sd["state"]) == len(params)
        q.put((rank, ok and sd_ok, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_zero1_matches_plain_adamw():
    world, port = 2, free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_zero_worker, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


def _zero_resume_worker(rank, world, port, q):
    try:
        _init(rank, world, port)
        import tempfile

        from relora_amd.parallel import ZeroRedundancyAdamW

        torch.manual_seed(0)
        model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
        opt = ZeroRedundancyAdamW(model.parameters(), lr=1e-2)
        for _ in range(3):
            loss = model(torch.randn(4, 8)).square().mean()
            loss.backward()
            opt.step()
            opt.zero_grad()

        # save: consolidate to rank 0, broadcast the dict to all (as a file
        # would be shared in practice)
        opt.consolidate_state_dict(to=0)
        payload = [opt.state_dict() if rank == 0 else None]
        dist.broadcast_object_list(payload, src=0)
        sd = payload[0]

        # fresh optimizer resumes and must hold identical shard states
        opt2 = ZeroRedundancyAdamW(model.parameters(), lr=1e-2)
        opt2.load_state_dict(sd)
        ok = True
        for p in opt.shard_params:
            a = opt.optim.state[p]
            b = opt2.optim.state[p]
            for k in ("exp_avg", "exp_avg_sq"):
                if not torch.allclose(a[k], b[k]):
                    ok = False
        # one more identical step on both must produce identical params
        g = [torch.randn_like(p) for p in model.parameters()]
        for p, gg in zip(model.parameters(), g):
            p.grad = gg.clone()
        opt2.step()
        q.put((rank, ok, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_zero1_consolidate_resume_roundtrip():
    port = free_port()
    q = mp.get_context("spawn").Queue()
    ps = [mp.get_context("spawn").Process(target=_zero_resume_worker, args=(r, 2, port, q))
          for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


def _world4_worker(rank, world, port, q):
    """DDP averaging + ZeRO-1 at world_size=4 with UNEVEN shards (5 params
    over 4 ranks) — the multi-GPU shape the driver's 8-GPU scaling run
    exercises, minus the hardware."""
    try:
        _init(rank, world, port)
        from relora_amd.parallel import DistributedModel, ZeroRedundancyAdamW
        from relora_amd.ops.optim import AdamW

        torch.manual_seed(0)
        model = torch.nn.Sequential(
            torch.nn.Linear(8, 12), torch.nn.ReLU(),
            torch.nn.Linear(12, 6, bias=True), torch.nn.Linear(6, 4, bias=False),
        )
        wrapped = DistributedModel(model, bucket_cap_mb=0.0001)
        params = [p for p in model.parameters() if p.requires_grad]
        opt = ZeroRedundancyAdamW(params, lr=1e-2, betas=(0.9, 0.999),
                                  weight_decay=0.01)

        ref = torch.nn.Sequential(
            torch.nn.Linear(8, 12), torch.nn.ReLU(),
            torch.nn.Linear(12, 6, bias=True), torch.nn.Linear(6, 4, bias=False),
        )
        ref.load_state_dict(model.state_dict())
        ref_opt = AdamW(list(ref.parameters()), lr=1e-2, betas=(0.9, 0.999),
                        weight_decay=0.01)

        torch.manual_seed(7)
        for it in range(3):
            x = torch.randn(world * 2, 8)
            y = torch.randn(world * 2, 4)
            xb, yb = x[rank * 2: rank * 2 + 2], y[rank * 2: rank * 2 + 2]
            wrapped.set_gradient_sync(True)
            torch.nn.functional.mse_loss(wrapped(xb), yb).backward()
            wrapped.finish_gradient_sync()
            opt.step()
            wrapped.zero_grad_buffers()

            # oracle: mean over the 4 per-rank losses on one process
            ref_loss = sum(
                torch.nn.functional.mse_loss(ref(x[r * 2: r * 2 + 2]),
                                             y[r * 2: r * 2 + 2])
                for r in range(world)) / world
            ref_loss.backward()
            ref_opt.step()
            ref_opt.zero_grad()

        ok = all(torch.allclose(p, rp, atol=1e-5)
                 for p, rp in zip(model.parameters(), ref.parameters()))
        opt.consolidate_state_dict()
        if rank == 0:
            sd = opt.state_dict()
            ok = ok and len(sd["state"]) == len(params)
        q.put((rank, ok, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_ddp_zero1_world4():
    world, port = 4, free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_world4_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


def _zero_reset_worker(rank, world, port, q):
    """ReLoRA's optimizer_reset against a REAL sharded ZeRO optimizer
    (2 processes): pruning must hit the rank-local shard states through the
    `.optim.state` surface (the reference's documented ZeRO quirk,
    training_utils.py:267-364) and training must continue."""
    try:
        _init(rank, world, port)
        from relora_amd import training_utils
        from relora_amd.parallel import DistributedModel, ZeroRedundancyAdamW

        torch.manual_seed(0)
        model = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Linear(32, 8))
        wrapped = DistributedModel(model)
        params = [p for p in model.parameters() if p.requires_grad]
        opt = ZeroRedundancyAdamW(params, lr=1e-2, betas=(0.9, 0.999))

        torch.manual_seed(5)
        for _ in range(3):
            x = torch.randn(4, 16)
            wrapped.set_gradient_sync(True)
            wrapped(x).pow(2).mean().backward()
            wrapped.finish_gradient_sync()
            opt.step()
            wrapped.zero_grad_buffers()

        # states are dense before reset on the owning rank
        owned = [p for p in params if p in opt.optim.state
                 and "exp_avg" in opt.optim.state[p]]
        assert owned, "rank owns no shard states"
        pre_zero = [float((opt.optim.state[p]["exp_avg"] == 0).float().mean())
                    for p in owned]

        training_utils.optimizer_reset(
            opt, reset_params=params,
            optimizer_state_keys=["exp_avg", "exp_avg_sq"],
            reset_optimizer_on_relora=False,
            optimizer_random_pruning=0.0,
            optimizer_magnitude_pruning=0.9,
        )
        post_zero = [float((opt.optim.state[p]["exp_avg"] == 0).float().mean())
                     for p in owned]
        ok = all(b > a and b >= 0.85 for a, b in zip(pre_zero, post_zero))

        # loop keeps running after the reset
        x = torch.randn(4, 16)
        wrapped.set_gradient_sync(True)
        wrapped(x).pow(2).mean().backward()
        wrapped.finish_gradient_sync()
        opt.step()
        q.put((rank, ok, None if ok else (pre_zero, post_zero)))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_zero_optimizer_reset_on_shards():
    world, port = 2, free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_zero_reset_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


# ---------------------------------------------------------------------------
# ZeRO-1 checkpoint cross-compat with torch.distributed ZeroRedundancyOptimizer
# (reference saves optimizer.pt via ZRO.consolidate_state_dict + state_dict —
# torchrun_main.py:204-218; both formats are position-indexed torch dicts)
# ---------------------------------------------------------------------------


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.Tanh(), torch.nn.Linear(32, 8))


def _zero_crosscompat_worker(rank, world, port, q):
    try:
        _init(rank, world, port)
        from torch.distributed.optim import ZeroRedundancyOptimizer

        from relora_amd.parallel.zero import ZeroRedundancyAdamW as ZeroAdamW

        # --- reference-style ZRO produces the checkpoint -------------------
        ref_model = _make_model()
        ref_opt = ZeroRedundancyOptimizer(
            ref_model.parameters(), optimizer_class=torch.optim.AdamW, lr=1e-3)
        for _ in range(3):
            ref_model(torch.randn(4, 16)).sum().backward()
            ref_opt.step()
            ref_opt.zero_grad()
        ref_opt.consolidate_state_dict(to=0)
        sd = ref_opt.state_dict() if rank == 0 else None
        holder = [sd]
        torch.distributed.broadcast_object_list(holder, src=0)
        sd = holder[0]

        # --- our ZeRO-1 loads it -------------------------------------------
        model = _make_model()
        opt = ZeroAdamW(model.parameters(), lr=1e-3)
        opt.load_state_dict(sd)
        ok = True
        for i, p in enumerate(opt.all_params):
            if opt.owner[i] != rank:
                continue
            st = opt.optim.state.get(p, {})
            ok = ok and torch.allclose(st["exp_avg"], sd["state"][i]["exp_avg"])
            ok = ok and torch.allclose(st["exp_avg_sq"], sd["state"][i]["exp_avg_sq"])

        # --- and round-trips back into a torch ZRO / plain AdamW ------------
        opt.consolidate_state_dict(to=0)
        ours = opt.state_dict() if rank == 0 else None
        if rank == 0:
            plain = torch.optim.AdamW(_make_model().parameters(), lr=1e-3)
            plain.load_state_dict(ours)  # torch validates the format
            for i in sd["state"]:
                ok = ok and torch.allclose(ours["state"][i]["exp_avg"],
                                           sd["state"][i]["exp_avg"])
        q.put((rank, bool(ok)))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, f"{e}\n{traceback.format_exc()}"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_zero1_crosscompat_with_torch_zero_redundancy():
    world, port = 2, free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_zero_crosscompat_worker, args=(r, world, port, q))
             for r in range(world)]
    [p.start() for p in procs]
    results = [q.get(timeout=180) for _ in range(world)]
    [p.join(timeout=60) for p in procs]
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"
