"""bench.py contract smoke: the exact multi-rank invocation the driver
uses must produce one valid JSON line (CPU/gloo mode, tiny model)."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def _run_bench(nproc, extra):
    env = dict(os.environ, FENGSHEN_BENCH_CPU="1",
               FENGSHEN_AMD_FORCE_EAGER="1")
    port = str(29000 + (os.getpid() % 900) + nproc)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
           "--master-port", port, os.path.join(ROOT, "bench.py"),
           "--model", "llama-tiny", "--micro_batch", "2", "--seq_len", "64",
           "--steps", "2", "--warmup", "1"] + extra
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         cwd=ROOT, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


@pytest.mark.parametrize("nproc,extra,parallelism", [
    (2, ["--zero_stage", "3"], "zero3_dp2"),
    (2, ["--zero_stage", "2", "--tp", "2"], "zero2_dp1_tp2"),
    # the driver's N=4 scaling point and the BASELINE config-4 topology
    (4, ["--zero_stage", "3"], "zero3_dp4"),
    (4, ["--zero_stage", "3", "--tp", "2"], "zero3_dp2_tp2"),
    # the driver's N=8 headline point
    (8, ["--zero_stage", "3"], "zero3_dp8"),
])
def test_bench_multirank_json(nproc, extra, parallelism):
    res = _run_bench(nproc, extra)
    assert res["n_gpus"] == nproc
    assert res["config"]["parallelism"] == parallelism
    assert res["value"] > 0 and res["ms_per_step"] > 0
    assert res["dtype"] == "bf16" and res["data"] == "synthetic"
    assert res["scaling"] == "weak" and res["higher_is_better"] is True
