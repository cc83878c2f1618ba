"""ZeRO-3 numerics: sharded params + gather-on-forward across 2 gloo ranks
must match a single-process AdamW run with the same decay split."""
import torch
import torch.nn as nn

from tests.distributed_utils import run_distributed


class Tiny(nn.Module):
    """Two 'layer' submodules + embedding/head so unit partitioning kicks in."""

    def __init__(self, d=32, seed=3):
        super().__init__()
        torch.manual_seed(seed)
        self.emb = nn.Linear(16, d)
        self.l1 = nn.Sequential(nn.Linear(d, d), nn.GELU(), nn.LayerNorm(d))
        self.l2 = nn.Sequential(nn.Linear(d, d), nn.GELU(), nn.LayerNorm(d))
        self.head = nn.Linear(d, 4)

    def forward(self, x):
        return self.head(self.l2(self.l1(self.emb(x))))


def _data(seed=11, n=64):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, 16, generator=g), torch.randn(n, 4, generator=g)


def _oracle(steps=5, lr=1e-2, wd=0.01):
    from fengshen_amd.parallel.zero3 import _is_no_decay
    model = Tiny()
    decay, nodecay = [], []
    for n_, p in model.named_parameters():
        (nodecay if _is_no_decay(n_, p) else decay).append(p)
    opt = torch.optim.AdamW(
        [{"params": decay, "weight_decay": wd},
         {"params": nodecay, "weight_decay": 0.0}],
        lr=lr, betas=(0.9, 0.999), eps=1e-8)
    x, y = _data()
    for _ in range(steps):
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
    return {k: v.detach().clone() for k, v in model.state_dict().items()}


def _zero3_worker(rank, world_size, steps=5, lr=1e-2, wd=0.01):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero3 import Zero3Engine

    init_distributed(backend="gloo")
    model = Tiny()
    eng = Zero3Engine(model, lr=lr, betas=(0.9, 0.999), eps=1e-8,
                      weight_decay=wd, unit_classes=(nn.Sequential,))
    x, y = _data()
    n = x.shape[0] // world_size
    xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    for _ in range(steps):
        loss = torch.nn.functional.mse_loss(model(xs), ys)
        eng.zero_grad()
        loss.backward()
        eng.step()
    with eng.gathered_params():
        out = {k: v.detach().clone() for k, v in model.state_dict().items()}
    dist.destroy_process_group()
    return out


def test_zero3_matches_adamw_world2():
    ref = _oracle()
    results = run_distributed(_zero3_worker, world_size=2)
    for sd in results:
        for k, v in ref.items():
            assert torch.allclose(v, sd[k], atol=2e-5, rtol=1e-4), (
                k, (v - sd[k]).abs().max().item())


def test_zero3_single_process():
    ref = _oracle(steps=3)
    from fengshen_amd.parallel.zero3 import Zero3Engine
    model = Tiny()
    eng = Zero3Engine(model, lr=1e-2, weight_decay=0.01,
                      unit_classes=(nn.Sequential,))
    x, y = _data()
    for _ in range(3):
        loss = torch.nn.functional.mse_loss(model(x), y)
        eng.zero_grad()
        loss.backward()
        eng.step()
    with eng.gathered_params():
        sd = model.state_dict()
        for k, v in _oracle(steps=3).items():
            assert torch.allclose(v, sd[k], atol=2e-5, rtol=1e-4), k


def _zero3_mem_worker(rank, world_size):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero3 import Zero3Engine

    init_distributed(backend="gloo")
    model = Tiny(d=64)
    total_numel = sum(p.numel() for p in model.parameters())
    eng = Zero3Engine(model, lr=1e-3, unit_classes=(nn.Sequential,))
    # after construction params are stubs; resident = shards only
    resident = sum(u.shard.numel() for u in eng.units)
    stubbed = all(p.numel() == 0 for u in eng.units for p in u.params)
    dist.destroy_process_group()
    return {"total": total_numel, "resident": resident, "stubbed": stubbed}


def test_zero3_actually_shards():
    results = run_distributed(_zero3_mem_worker, world_size=2)
    for r in results:
        assert r["stubbed"]
        # padded shards: roughly half the params (+ padding)
        assert r["resident"] < r["total"] * 0.75


def _zero3_accum_worker(rank, world_size):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero3 import Zero3Engine

    init_distributed(backend="gloo")
    model = Tiny()
    eng = Zero3Engine(model, lr=1e-2, weight_decay=0.01,
                      unit_classes=(nn.Sequential,))
    x, y = _data()
    n = x.shape[0] // world_size
    xs, ys = x[rank * n:(rank + 1) * n], y[rank * n:(rank + 1) * n]
    half = n // 2
    for _ in range(3):
        eng.zero_grad()
        for micro in range(2):
            eng.set_sync(micro == 1)
            xm = xs[micro * half:(micro + 1) * half]
            ym = ys[micro * half:(micro + 1) * half]
            loss = torch.nn.functional.mse_loss(model(xm), ym) / 2
            loss.backward()
        eng.step()
    with eng.gathered_params():
        out = {k: v.detach().clone() for k, v in model.state_dict().items()}
    dist.destroy_process_group()
    return out


def test_zero3_grad_accumulation():
    ref = _oracle(steps=3)
    results = run_distributed(_zero3_accum_worker, world_size=2)
    for sd in results:
        for k, v in ref.items():
            assert torch.allclose(v, sd[k], atol=2e-5, rtol=1e-4), k


class TiedBlock(nn.Module):
    """Unit-class module that CONTAINS the embedding later tied to the head."""

    def __init__(self, d, vocab):
        super().__init__()
        self.emb = nn.Embedding(vocab, d)
        self.ln = nn.LayerNorm(d)

    def forward(self, ids):
        return self.ln(self.emb(ids))


class TiedModel(nn.Module):
    """Embedding inside a unit module, tied into an outside LM head — the
    hazardous ZeRO-3 case: the head's forward must not see released storage.
    The engine must place the tied weight in the resident "(rest)" unit."""

    def __init__(self, d=32, vocab=50, seed=5):
        super().__init__()
        torch.manual_seed(seed)
        self.block = TiedBlock(d, vocab)
        self.mid = nn.Sequential(nn.Linear(d, d), nn.GELU())
        self.head = nn.Linear(d, vocab, bias=False)
        self.head.weight = self.block.emb.weight  # tie

    def forward(self, ids):
        return self.head(self.mid(self.block(ids)))


def _tied_data(seed=13, n=16, vocab=50):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, vocab, (n, 8), generator=g)
    return ids


def _tied_oracle(steps=4, lr=1e-2, wd=0.01):
    from fengshen_amd.parallel.zero3 import _is_no_decay
    model = TiedModel()
    decay, nodecay = [], []
    seen = set()
    for n_, p in model.named_parameters():
        if id(p) in seen:
            continue
        seen.add(id(p))
        (nodecay if _is_no_decay(n_, p) else decay).append(p)
    opt = torch.optim.AdamW(
        [{"params": decay, "weight_decay": wd},
         {"params": nodecay, "weight_decay": 0.0}],
        lr=lr, betas=(0.9, 0.999), eps=1e-8)
    ids = _tied_data()
    tgt = ids.roll(-1, dims=1)
    for _ in range(steps):
        logits = model(ids)
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), tgt.reshape(-1))
        opt.zero_grad()
        loss.backward()
        opt.step()
    return {k: v.detach().clone() for k, v in model.state_dict().items()}


def _tied_worker(rank, world_size, steps=4, lr=1e-2, wd=0.01):
    import torch.distributed as dist
    from fengshen_amd.parallel.groups import init_distributed
    from fengshen_amd.parallel.zero3 import Zero3Engine

    init_distributed(backend="gloo")
    model = TiedModel()
    eng = Zero3Engine(model, lr=lr, betas=(0.9, 0.999), eps=1e-8,
                      weight_decay=wd,
                      unit_classes=(TiedBlock, nn.Sequential))
    # tied weight must NOT be claimed by the TiedBlock unit
    for u in eng.units:
        if u.name != "(rest)":
            assert all(p is not model.head.weight for p in u.params), u.name
    assert any(p is model.head.weight
               for u in eng.units if u.name == "(rest)" for p in u.params)

    ids = _tied_data()
    tgt = ids.roll(-1, dims=1)
    n = ids.shape[0] // world_size
    ids_r, tgt_r = ids[rank * n:(rank + 1) * n], tgt[rank * n:(rank + 1) * n]
    for _ in range(steps):
        logits = model(ids_r)
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), tgt_r.reshape(-1))
        eng.zero_grad()
        loss.backward()
        eng.step()
    with eng.gathered_params():
        out = {k: v.detach().clone() for k, v in model.state_dict().items()}
    dist.destroy_process_group()
    return out


def test_zero3_tied_parameters_world2():
    ref = _tied_oracle()
    results = run_distributed(_tied_worker, world_size=2)
    for sd in results:
        for k, v in ref.items():
            assert torch.allclose(v, sd[k], atol=2e-5, rtol=1e-4), (
                k, (v - sd[k]).abs().max().item())
