"""DDP GradReducer (non-Adam path): gradient averaging across ranks
matches a single-process full-batch run."""
import torch
import torch.nn as nn

from tests.distributed_utils import run_distributed


def _build(seed=3):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 1))


def _data(n=32, d=8, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, d, generator=g)
    w = torch.arange(1, d + 1, dtype=torch.float32)
    return x, x @ w


def _ddp_sgd_worker(rank, world_size, _):
    import torch.distributed as dist

    from fengshen_amd.parallel.ddp import GradReducer
    from fengshen_amd.parallel.groups import init_distributed

    init_distributed(backend="gloo")
    model = _build()
    reducer = GradReducer(model)
    opt = torch.optim.SGD(model.parameters(), lr=1e-2)
    x, y = _data()
    n = x.shape[0] // world_size
    xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    for _step in range(4):
        loss = torch.nn.functional.mse_loss(model(xs).squeeze(-1), ys)
        opt.zero_grad()
        loss.backward()   # hooks launch bucket all-reduces
        reducer.finalize()
        opt.step()
    out = [p.detach().clone() for p in model.parameters()]
    dist.destroy_process_group()
    return out


def test_grad_reducer_matches_full_batch():
    model = _build()
    opt = torch.optim.SGD(model.parameters(), lr=1e-2)
    x, y = _data()
    for _step in range(4):
        loss = torch.nn.functional.mse_loss(model(x).squeeze(-1), y)
        opt.zero_grad()
        loss.backward()
        opt.step()

    results = run_distributed(_ddp_sgd_worker, world_size=2, args=(None,))
    for rank_params in results:
        for p_ref, p_ddp in zip(model.parameters(), rank_params):
            assert torch.allclose(p_ref, p_ddp, atol=1e-5), \
                (p_ref - p_ddp).abs().max()
