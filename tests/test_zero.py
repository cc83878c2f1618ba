"""Native ZeRO numerics: stage 0/1/2 across 2 gloo ranks must match a
single-process full-batch AdamW run bitwise-ish (fp32)."""
import pytest
import torch
import torch.nn as nn

from tests.distributed_utils import run_distributed


def _build_model(seed=3):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(16, 64), nn.GELU(), nn.Linear(64, 64), nn.GELU(),
        nn.Linear(64, 4))


def _data(seed=11, n=64):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 16, generator=g)
    y = torch.randn(n, 4, generator=g)
    return x, y


def _reference_run(steps=5, lr=1e-2):
    """Single-process full-batch AdamW (the oracle)."""
    model = _build_model()
    opt = torch.optim.AdamW(model.parameters(), lr=lr, betas=(0.9, 0.999),
                            eps=1e-8, weight_decay=0.01)
    x, y = _data()
    for _ in range(steps):
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def _zero_worker(rank, world_size, stage, steps=5, lr=1e-2):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero import ZeroOptimizer

    init_distributed(backend="gloo")
    model = _build_model()
    opt = ZeroOptimizer(model.parameters(), stage=stage, lr=lr,
                        betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01,
                        bucket_numel=2000)  # tiny buckets -> exercise bucketing
    x, y = _data()
    n = x.shape[0] // world_size
    xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    for _ in range(steps):
        loss = torch.nn.functional.mse_loss(model(xs), ys)
        opt.zero_grad()
        loss.backward()
        opt.step()
    out = [p.detach().clone() for p in model.parameters()]
    dist.destroy_process_group()
    return out


@pytest.mark.parametrize("stage", [0, 1, 2])
def test_zero_matches_adamw(stage):
    ref = _reference_run()
    results = run_distributed(_zero_worker, world_size=2, args=(stage,))
    for rank_params in results:
        assert len(rank_params) == len(ref)
        for p_ref, p_zero in zip(ref, rank_params):
            assert torch.allclose(p_ref, p_zero.float(), atol=1e-5, rtol=1e-4), (
                f"stage {stage}: max diff "
                f"{(p_ref - p_zero.float()).abs().max().item()}")


def _zero_accum_worker(rank, world_size, stage):
    """Grad accumulation: 2 micro-batches/step must equal 1 full batch/step."""
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero import ZeroOptimizer

    init_distributed(backend="gloo")
    model = _build_model()
    opt = ZeroOptimizer(model.parameters(), stage=stage, lr=1e-2,
                        weight_decay=0.01, bucket_numel=2000)
    x, y = _data()
    n = x.shape[0] // world_size
    xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    half = n // 2
    for _ in range(3):
        opt.zero_grad()
        for micro in range(2):
            opt.set_sync(micro == 1)
            xm = xs[micro * half:(micro + 1) * half]
            ym = ys[micro * half:(micro + 1) * half]
            loss = torch.nn.functional.mse_loss(model(xm), ym) / 2
            loss.backward()
        opt.step()
    out = [p.detach().clone() for p in model.parameters()]
    dist.destroy_process_group()
    return out


def test_zero_grad_accumulation():
    ref = _reference_run(steps=3)
    results = run_distributed(_zero_accum_worker, world_size=2, args=(2,))
    for rank_params in results:
        for p_ref, p_zero in zip(ref, rank_params):
            assert torch.allclose(p_ref, p_zero.float(), atol=1e-5, rtol=1e-4)


def test_zero_single_process_stage0():
    """ZeroOptimizer degrades to fused AdamW when world_size == 1."""
    from fengshen_amd.parallel.zero import ZeroOptimizer
    ref = _reference_run()
    model = _build_model()
    opt = ZeroOptimizer(model.parameters(), stage=0, lr=1e-2, weight_decay=0.01)
    x, y = _data()
    for _ in range(5):
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
    for p_ref, p in zip(ref, model.parameters()):
        assert torch.allclose(p_ref, p, atol=1e-5, rtol=1e-4)


def test_zero_state_dict_roundtrip():
    from fengshen_amd.parallel.zero import ZeroOptimizer
    model = _build_model()
    opt = ZeroOptimizer(model.parameters(), stage=0, lr=1e-2)
    x, y = _data()
    for _ in range(2):
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
    sd = opt.state_dict()
    model2 = _build_model(seed=99)
    opt2 = ZeroOptimizer(model2.parameters(), stage=0, lr=1e-2)
    opt2.load_state_dict(sd)
    # load refreshes model2's params from the master shards
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def _zero_offload_worker(rank, world_size, stage):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero import ZeroOptimizer

    init_distributed(backend="gloo")
    model = _build_model()
    opt = ZeroOptimizer(model.parameters(), stage=stage, lr=1e-2,
                        betas=(0.9, 0.999), eps=1e-8, weight_decay=0.01,
                        bucket_numel=2000, cpu_offload=True)
    assert all(not b.master_shard.is_cuda for b in opt.buckets)
    x, y = _data()
    n = x.shape[0] // world_size
    xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    for _ in range(3):
        loss = torch.nn.functional.mse_loss(model(xs), ys)
        opt.zero_grad()
        loss.backward()
        opt.step()
    out = [p.detach().clone() for p in model.parameters()]
    dist.destroy_process_group()
    return out


@pytest.mark.parametrize("stage", [1, 2])
def test_zero_offload_matches_adamw(stage):
    """ZeRO-offload (host-RAM optimizer states) is numerically identical
    to the on-device path (reference: deepspeed offload_optimizer)."""
    ref = _reference_run(steps=3)
    results = run_distributed(_zero_offload_worker, world_size=2,
                              args=(stage,))
    for rank_params in results:
        for p_ref, p_zero in zip(ref, rank_params):
            assert torch.allclose(p_ref, p_zero.float(), atol=1e-5,
                                  rtol=1e-4)


def test_parse_strategy_offload():
    from fengshen_amd.trainer.strategies import parse_strategy
    s = parse_strategy("zero2_offload")
    assert s == {"kind": "zero", "stage": 2, "cpu_offload": True}
    assert parse_strategy("zero2")["cpu_offload"] is False
