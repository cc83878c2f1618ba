"""Example apps stay runnable (synthetic smoke mode, subprocess)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def _run(rel, *extra):
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, rel), *extra],
        capture_output=True, text=True, timeout=300, cwd=ROOT,
        env=dict(os.environ, FENGSHEN_AMD_FORCE_EAGER="1"))
    assert out.returncode == 0, (rel, out.stderr[-1500:])
    return out.stdout


@pytest.mark.parametrize("rel,extra", [
    ("examples/tcbert/example_tcbert.py", ()),
    ("examples/uniex/example_uniex.py", ()),
    ("examples/ppvae_gavae/latent_plugins.py", ("--steps", "5")),
    ("examples/davae_generate/generate_davae.py",
     ("--n", "2", "--seq_len", "8")),
    ("examples/transfo_xl_denoise/generate.py", ("--seq_len", "28")),
    ("examples/fastdemo/qa_demo.py", ("--smoke",)),
    ("examples/disco_project/clip_guided_generate.py", ("--steps", "4")),
])
def test_example_smokes(rel, extra):
    _run(rel, *extra)


def test_training_example_one_step():
    out = _run("examples/clue_sim/finetune_clue_sim.py",
               "--max_steps", "1", "--strategy", "ddp")
    assert "train_loss" in out
