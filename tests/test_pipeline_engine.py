"""1F1B pipeline engine numerics: stage-split model over a gloo pipeline
group must reproduce single-process loss and gradients exactly."""
import torch
import torch.nn as nn

from tests.distributed_utils import run_distributed


def _layers(d=16, n=6, seed=3):
    torch.manual_seed(seed)
    return [nn.Sequential(nn.Linear(d, d), nn.Tanh()) for _ in range(n)]


def _data(b=8, d=16, seed=11):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(b, d, generator=g),
            torch.randn(b, d, generator=g))


def _single_process_reference(num_micro=4):
    layers = _layers()
    model = nn.Sequential(*layers)
    x, y = _data()
    mbs = x.chunk(num_micro)
    tgs = y.chunk(num_micro)
    total = 0.0
    for mb, tg in zip(mbs, tgs):
        loss = torch.nn.functional.mse_loss(model(mb), tg)
        (loss / num_micro).backward()
        total += float(loss)
    grads = [p.grad.clone() for p in model.parameters()]
    return total / num_micro, grads


def _pipeline_worker(rank, world_size, num_micro=4):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import (
        init_distributed, initialize_model_parallel)
    from fengshen_amd.parallel.pipeline import (
        PipelineEngine, split_module_for_pipeline)

    init_distributed(backend="gloo")
    initialize_model_parallel(1, pipeline_model_parallel_size=world_size)
    layers = _layers()
    stage = split_module_for_pipeline(layers, world_size, rank)
    x, y = _data()
    num = x.shape[0] // num_micro
    eng = PipelineEngine(
        stage, lambda out, tg: torch.nn.functional.mse_loss(out, tg),
        num_microbatches=num_micro, act_shape=(num, 16),
        act_dtype=torch.float32)
    mbs = list(x.chunk(num_micro)) if rank == 0 else None
    tgs = list(y.chunk(num_micro)) if rank == world_size - 1 else None
    loss = eng.train_batch(mbs, tgs)
    grads = [p.grad.clone() for p in stage.parameters()]
    dist.destroy_process_group()
    return {"rank": rank, "loss": loss, "grads": grads,
            "n_stage_layers": len(stage)}


def _check(world_size):
    ref_loss, ref_grads = _single_process_reference()
    results = run_distributed(_pipeline_worker, world_size=world_size)
    results.sort(key=lambda r: r["rank"])
    # last stage reports the averaged loss
    assert abs(results[-1]["loss"] - ref_loss) < 1e-5
    assert all(r["loss"] is None for r in results[:-1])
    # stage grads concatenated == single-process grads
    flat = [g for r in results for g in r["grads"]]
    assert len(flat) == len(ref_grads)
    for got, want in zip(flat, ref_grads):
        assert torch.allclose(got, want, atol=1e-6), \
            (got - want).abs().max()


def test_pipeline_1f1b_world2():
    _check(2)


def test_pipeline_1f1b_world3():
    _check(3)


def test_pipeline_split_contiguous():
    from fengshen_amd.parallel.pipeline import split_module_for_pipeline
    layers = _layers(n=7)
    s0 = split_module_for_pipeline(layers, 3, 0)
    s1 = split_module_for_pipeline(layers, 3, 1)
    s2 = split_module_for_pipeline(layers, 3, 2)
    assert len(s0) == 3 and len(s1) == 3 and len(s2) == 1


def test_pipeline_1f1b_world4():
    """4-stage 1F1B (the driver's N=4 shape): gradient-equivalence vs
    single-process reference, warmup depth 3 on stage 0."""
    _check(4)
