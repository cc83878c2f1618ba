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


class _TPBlock(torch.nn.Module):
    """Column(d->2d, gelu) -> Row(2d->d): the minimal TP layer pair."""

    def __init__(self, d):
        super().__init__()
        from fengshen_amd.parallel.layers import (
            ColumnParallelLinear, RowParallelLinear)
        self.col = ColumnParallelLinear(d, 2 * d, bias=True,
                                        gather_output=False,
                                        dtype=torch.float32)
        self.row = RowParallelLinear(2 * d, d, bias=True,
                                     input_is_parallel=True,
                                     dtype=torch.float32)

    def forward(self, x):
        return self.row(torch.nn.functional.gelu(self.col(x)))


def _tp_pp_worker(rank, world_size):
    """world 4 = tp2 x pp2: TP blocks inside pipeline stages must match
    the dense single-process model's loss and (sharded) gradients."""
    import torch.distributed as dist
    from fengshen_amd.parallel import groups
    from fengshen_amd.parallel.groups import (
        init_distributed, initialize_model_parallel)
    from fengshen_amd.parallel.pipeline import PipelineEngine

    d, num_micro = 16, 2
    init_distributed(backend="gloo")
    initialize_model_parallel(2, pipeline_model_parallel_size=2)
    tp_rank = groups.get_tensor_model_parallel_rank()
    pp_rank = groups.get_pipeline_model_parallel_rank()

    # dense reference weights, identical on every rank
    torch.manual_seed(21)
    ref = [torch.nn.ModuleDict({
        "col": torch.nn.Linear(d, 2 * d), "row": torch.nn.Linear(2 * d, d)})
        for _ in range(2)]
    x, y = _data(b=4, d=d, seed=23)

    blk = _TPBlock(d)
    r = ref[pp_rank]
    per = 2 * d // 2  # column shard rows / row shard cols
    with torch.no_grad():
        blk.col.weight.copy_(r["col"].weight[tp_rank * per:(tp_rank + 1) * per])
        blk.col.bias.copy_(r["col"].bias[tp_rank * per:(tp_rank + 1) * per])
        blk.row.weight.copy_(r["row"].weight[:, tp_rank * per:(tp_rank + 1) * per])
        blk.row.bias.copy_(r["row"].bias)

    eng = PipelineEngine(
        blk, lambda out, tg: torch.nn.functional.mse_loss(out, tg),
        num_microbatches=num_micro, act_shape=(x.shape[0] // num_micro, d),
        act_dtype=torch.float32)
    mbs = list(x.chunk(num_micro)) if pp_rank == 0 else None
    tgs = list(y.chunk(num_micro)) if pp_rank == 1 else None
    loss = eng.train_batch(mbs, tgs)

    # dense reference forward/backward
    def dense(x):
        for m in ref:
            x = m["row"](torch.nn.functional.gelu(m["col"](x)))
        return x
    total = 0.0
    for mb, tg in zip(x.chunk(num_micro), y.chunk(num_micro)):
        l = torch.nn.functional.mse_loss(dense(mb), tg)
        (l / num_micro).backward()
        total += float(l) / num_micro
    rg = ref[pp_rank]
    col_g = rg["col"].weight.grad[tp_rank * per:(tp_rank + 1) * per]
    row_g = rg["row"].weight.grad[:, tp_rank * per:(tp_rank + 1) * per]
    err = max(float((blk.col.weight.grad - col_g).abs().max()),
              float((blk.row.weight.grad - row_g).abs().max()))
    dist.destroy_process_group()
    return {"pp": pp_rank, "tp": tp_rank, "loss": loss, "ref_loss": total,
            "err": err}


def test_pipeline_tp2_pp2_grads():
    results = run_distributed(_tp_pp_worker, world_size=4)
    for r in results:
        assert r["err"] < 1e-5, r
        if r["loss"] is not None:  # last pp stage reports loss
            assert abs(r["loss"] - r["ref_loss"]) < 1e-5, r


def _dp_pp_worker(rank, world_size):
    """world 4 = dp2 x pp2: pipeline stages wrapped in the DP GradReducer
    must produce grads equal to the dense model over the FULL batch."""
    import torch.distributed as dist
    from fengshen_amd.parallel import groups
    from fengshen_amd.parallel.ddp import GradReducer
    from fengshen_amd.parallel.groups import (
        init_distributed, initialize_model_parallel)
    from fengshen_amd.parallel.pipeline import (
        PipelineEngine, split_module_for_pipeline)

    num_micro = 2
    init_distributed(backend="gloo")
    initialize_model_parallel(1, pipeline_model_parallel_size=2)
    dp_rank = groups.get_data_parallel_rank()
    pp_rank = groups.get_pipeline_model_parallel_rank()

    layers = _layers(n=4, seed=31)
    stage = split_module_for_pipeline(layers, 2, pp_rank)
    reducer = GradReducer(stage, process_group=groups.get_data_parallel_group())
    x, y = _data(b=8, seed=33)
    # each DP replica consumes its half of the batch
    xr, yr = x.chunk(2)[dp_rank], y.chunk(2)[dp_rank]
    eng = PipelineEngine(
        stage, lambda out, tg: torch.nn.functional.mse_loss(out, tg),
        num_microbatches=num_micro, act_shape=(xr.shape[0] // num_micro, 16),
        act_dtype=torch.float32)
    reducer.set_sync(False)  # accumulate across microbatches
    eng.train_batch(list(xr.chunk(num_micro)) if pp_rank == 0 else None,
                    list(yr.chunk(num_micro)) if pp_rank == 1 else None)
    reducer.set_sync(True)
    reducer.finalize()

    # dense reference over the FULL batch (mean of the two replicas)
    model = torch.nn.Sequential(*_layers(n=4, seed=31))
    for mb, tg in zip(x.chunk(2 * num_micro), y.chunk(2 * num_micro)):
        (torch.nn.functional.mse_loss(model(mb), tg) / num_micro / 2).backward()
    ref = torch.nn.Sequential(*split_module_for_pipeline(
        list(model), 2, pp_rank))
    err = max(float((p.grad - q.grad).abs().max())
              for p, q in zip(stage.parameters(), ref.parameters()))
    dist.destroy_process_group()
    return {"pp": pp_rank, "dp": dp_rank, "err": err}


def test_pipeline_dp2_pp2_grad_average():
    results = run_distributed(_dp_pp_worker, world_size=4)
    for r in results:
        assert r["err"] < 1e-5, r


def _dp_tp_pp_worker(rank, world_size):
    """world 8 = dp2 x tp2 x pp2 — the full 3D composition: TP blocks
    inside pipeline stages, DP gradient averaging across replicas, all
    exact vs the dense full-batch reference."""
    import torch.distributed as dist
    from fengshen_amd.parallel import groups
    from fengshen_amd.parallel.ddp import GradReducer
    from fengshen_amd.parallel.groups import (
        init_distributed, initialize_model_parallel)
    from fengshen_amd.parallel.pipeline import PipelineEngine

    d, num_micro = 16, 2
    init_distributed(backend="gloo")
    initialize_model_parallel(2, pipeline_model_parallel_size=2)
    tp_rank = groups.get_tensor_model_parallel_rank()
    pp_rank = groups.get_pipeline_model_parallel_rank()
    dp_rank = groups.get_data_parallel_rank()

    torch.manual_seed(41)
    ref = [torch.nn.ModuleDict({
        "col": torch.nn.Linear(d, 2 * d), "row": torch.nn.Linear(2 * d, d)})
        for _ in range(2)]
    x, y = _data(b=8, d=d, seed=43)

    blk = _TPBlock(d)
    r = ref[pp_rank]
    per = d  # 2d // tp2
    with torch.no_grad():
        blk.col.weight.copy_(r["col"].weight[tp_rank * per:(tp_rank + 1) * per])
        blk.col.bias.copy_(r["col"].bias[tp_rank * per:(tp_rank + 1) * per])
        blk.row.weight.copy_(r["row"].weight[:, tp_rank * per:(tp_rank + 1) * per])
        blk.row.bias.copy_(r["row"].bias)
    reducer = GradReducer(blk, process_group=groups.get_data_parallel_group())

    xr, yr = x.chunk(2)[dp_rank], y.chunk(2)[dp_rank]
    eng = PipelineEngine(
        blk, lambda out, tg: torch.nn.functional.mse_loss(out, tg),
        num_microbatches=num_micro, act_shape=(xr.shape[0] // num_micro, d),
        act_dtype=torch.float32)
    reducer.set_sync(False)
    eng.train_batch(list(xr.chunk(num_micro)) if pp_rank == 0 else None,
                    list(yr.chunk(num_micro)) if pp_rank == 1 else None)
    reducer.set_sync(True)
    reducer.finalize()

    def dense(x):
        for m in ref:
            x = m["row"](torch.nn.functional.gelu(m["col"](x)))
        return x
    for mb, tg in zip(x.chunk(2 * num_micro), y.chunk(2 * num_micro)):
        (torch.nn.functional.mse_loss(dense(mb), tg) / num_micro / 2).backward()
    rg = ref[pp_rank]
    col_g = rg["col"].weight.grad[tp_rank * per:(tp_rank + 1) * per]
    row_g = rg["row"].weight.grad[:, tp_rank * per:(tp_rank + 1) * per]
    err = max(float((blk.col.weight.grad - col_g).abs().max()),
              float((blk.row.weight.grad - row_g).abs().max()),
              float((blk.row.bias.grad - rg["row"].bias.grad).abs().max()))
    dist.destroy_process_group()
    return {"dp": dp_rank, "tp": tp_rank, "pp": pp_rank, "err": err}


def test_pipeline_3d_dp2_tp2_pp2():
    results = run_distributed(_dp_tp_pp_worker, world_size=8)
    assert len(results) == 8
    for r in results:
        assert r["err"] < 1e-5, r


def _pp_zero2_worker(rank, world_size):
    """world 4 = dp2 x pp2 with ZeRO-2 sharded optimizer per stage:
    post-step parameters must equal a single-process model over the full
    batch stepped by the same (world-1) optimizer."""
    import torch.distributed as dist
    from fengshen_amd.parallel import groups
    from fengshen_amd.parallel.groups import (
        init_distributed, initialize_model_parallel)
    from fengshen_amd.parallel.pipeline import (
        PipelineEngine, split_module_for_pipeline)
    from fengshen_amd.parallel.zero import ZeroOptimizer

    num_micro = 2
    init_distributed(backend="gloo")
    initialize_model_parallel(1, pipeline_model_parallel_size=2)
    dp_rank = groups.get_data_parallel_rank()
    pp_rank = groups.get_pipeline_model_parallel_rank()

    layers = _layers(n=4, seed=51)
    stage = split_module_for_pipeline(layers, 2, pp_rank)
    opt = ZeroOptimizer(stage.parameters(), stage=2, lr=1e-2,
                        weight_decay=0.01,
                        process_group=groups.get_data_parallel_group())
    x, y = _data(b=8, seed=53)
    xr, yr = x.chunk(2)[dp_rank], y.chunk(2)[dp_rank]
    eng = PipelineEngine(
        stage, lambda out, tg: torch.nn.functional.mse_loss(out, tg),
        num_microbatches=num_micro, act_shape=(xr.shape[0] // num_micro, 16),
        act_dtype=torch.float32)
    opt.set_sync(False)  # accumulate across the 1F1B microbatches
    eng.train_batch(list(xr.chunk(num_micro)) if pp_rank == 0 else None,
                    list(yr.chunk(num_micro)) if pp_rank == 1 else None)
    opt.set_sync(True)
    opt.step()

    # single-process reference: full batch, same optimizer impl.
    # NOTE set_sync(False) during the accumulation backwards — with
    # sync on, the overlap hook fires after the FIRST micro-backward
    # and the async all-reduce races the later accumulations (the
    # documented contract: sync off during grad-accumulation steps).
    model = torch.nn.Sequential(*_layers(n=4, seed=51))
    ref_opt = ZeroOptimizer(model.parameters(), stage=2, lr=1e-2,
                            weight_decay=0.01, process_group=None)
    ref_opt.set_sync(False)
    for mb, tg in zip(x.chunk(2 * num_micro), y.chunk(2 * num_micro)):
        (torch.nn.functional.mse_loss(model(mb), tg) / num_micro / 2).backward()
    ref_opt.set_sync(True)
    ref_opt.step()
    ref_stage = split_module_for_pipeline(list(model), 2, pp_rank)
    err = max(float((p - q).abs().max())
              for p, q in zip(stage.parameters(), ref_stage.parameters()))
    dist.destroy_process_group()
    return {"pp": pp_rank, "dp": dp_rank, "err": err}


def test_pipeline_pp2_zero2_step():
    results = run_distributed(_pp_zero2_worker, world_size=4)
    for r in results:
        assert r["err"] < 1e-5, r
