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
    ("examples/randeng_reasoning/reasoning_generate.py",
     ("--max_out_seq", "10")),
    ("examples/longformer/longformer_mlm.py", ("--seq_len", "128")),
])
def test_example_smokes(rel, extra):
    _run(rel, *extra)


def test_training_example_one_step():
    out = _run("examples/clue_sim/finetune_clue_sim.py",
               "--max_steps", "1", "--strategy", "ddp")
    assert "train_loss" in out


@pytest.mark.parametrize("rel,extra", [
    ("examples/zen1_finetune/fengshen_sequence_level_ft_task.py",
     ("--max_steps", "2", "--precision", "fp32")),
    ("examples/zen1_finetune/fengshen_token_level_ft_task.py",
     ("--max_steps", "2", "--precision", "fp32")),
    ("examples/summary/seq2seq_summary.py",
     ("--max_steps", "2", "--precision", "fp32")),
    ("examples/clue1.1/run_clue_unimc.py",
     ("--max_steps", "2", "--precision", "fp32")),
    ("examples/unimc/finetune_unimc.py",
     ("--max_steps", "2", "--precision", "fp32")),
])
def test_new_example_smokes(rel, extra, tmp_path):
    _run(rel, *extra, "--default_root_dir", str(tmp_path))


def test_clue_converters_roundtrip(tmp_path):
    import json
    import subprocess
    src = tmp_path / "tnews.json"
    src.write_text(json.dumps(
        {"sentence": "球队赢得比赛", "label_desc": "news_sports",
         "id": 7}, ensure_ascii=False) + "\n")
    dst = tmp_path / "uni.jsonl"
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples/clue1.1/clue2unidata.py"),
         "--task", "tnews", "--input", str(src), "--output", str(dst)],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-500:]
    rec = json.loads(dst.read_text())
    assert rec["answer"] == "体育" and rec["choice"][rec["label"]] == "体育"


def test_launcher_scripts_exist_and_reference_real_files():
    import glob
    import re
    launchers = glob.glob(os.path.join(ROOT, "examples/*/run.sh"))
    assert len(launchers) >= 10
    for sh in launchers:
        body = open(sh).read()
        m = re.search(r"exec python (?:-m \S+ .*?)?(\S+\.py)", body)
        assert m, sh
        assert os.path.exists(os.path.join(os.path.dirname(sh),
                                           m.group(1))), (sh, m.group(1))


def test_rouge_score():
    from fengshen_amd.metric.rouge import RougeScore, rouge_l, rouge_n
    r = rouge_n(list("股市大涨"), list("股市大涨"), 1)
    assert r["fmeasure"] == 1.0
    r2 = rouge_l(list("股市涨"), list("股市大涨"))
    assert abs(r2["recall"] - 3 / 4) < 1e-9
    rs = RougeScore()
    rs.update(["股 市 涨"], ["股 市 大 涨"])
    out = rs.compute()
    assert 0 < out["rougeL_fmeasure"] < 1
    assert out["rouge1_precision"] == 1.0


@pytest.mark.parametrize("rel", [
    "examples/pretrain_erlangshen_bert/pretrain_erlangshen.py",
    "examples/ziya_llama/finetune_ziya_llama.py",
    "examples/classification/finetune_classification.py",
    "examples/sequence_tagging/finetune_tagging.py",
    "examples/finetune_bart_qg/finetune_bart_qg.py",
    "examples/mt5_summary/finetune_summary.py",
    "examples/translate/finetune_deltalm.py",
    "examples/wenzhong_qa/finetune_medicalQA.py",
    "examples/hubert/pretrain_hubert.py",
    "examples/deepVAE/pretrain_deep_vae.py",
    "examples/pegasus/pretrain_pegasus.py",
    "examples/clip_finetune/clip_finetune_flickr.py",
    "examples/pretrain_taiyi_clip/pretrain_clip.py",
    "examples/finetune_taiyi_stable_diffusion/finetune.py",
    "examples/pretrain_deberta_v2/pretrain_deberta.py",
    "examples/pretrain_bert/pretrain_bert_mmap.py",
    "examples/pretrain_t5/pretrain_randeng_t5.py",
    "examples/pretrain_randeng_bart/pretrain_bart.py",
    "examples/zen2_finetune/finetune_zen_classification.py",
    "examples/qa_t5/finetune_t5_qa.py",
])
def test_training_example_smokes(rel, tmp_path):
    """Every training app runs 2 steps on synthetic data (the reference's
    example-as-integration-test pattern)."""
    _run(rel, "--max_steps", "2", "--precision", "fp32",
         "--default_root_dir", str(tmp_path))
