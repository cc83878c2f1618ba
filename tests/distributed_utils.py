"""Spawn-based multi-process test harness (gloo on CPU, RCCL on GPU).

Pattern required by SURVEY.md §4: TP/ZeRO numerics exercised via gloo
world_size>1 on CPU here, and on a single MI355X via multi-process NCCL.
"""
import os
import pickle
import tempfile
import traceback

import torch.multiprocessing as mp


def _worker(rank, world_size, port, fn, args, result_dir):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("FENGSHEN_AMD_FORCE_EAGER", "1")
    try:
        out = fn(rank, world_size, *args)
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("ok", out), f)
    except Exception:
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("err", traceback.format_exc()), f)
        raise


def run_distributed(fn, world_size=2, args=(), timeout=180):
    """Run fn(rank, world_size, *args) in world_size processes; return
    list of per-rank results."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    with tempfile.TemporaryDirectory() as result_dir:
        ctx = mp.get_context("spawn")
        procs = []
        for r in range(world_size):
            p = ctx.Process(target=_worker,
                            args=(r, world_size, port, fn, args, result_dir))
            p.start()
            procs.append(p)
        for p in procs:
            p.join(timeout)
            if p.is_alive():
                for q in procs:
                    q.terminate()
                raise TimeoutError(f"distributed test timed out after {timeout}s")
        results = []
        for r in range(world_size):
            path = os.path.join(result_dir, f"rank{r}.pkl")
            if not os.path.exists(path):
                raise RuntimeError(f"rank {r} produced no result (crashed?)")
            with open(path, "rb") as f:
                status, payload = pickle.load(f)
            if status == "err":
                raise RuntimeError(f"rank {r} failed:\n{payload}")
            results.append(payload)
        return results
