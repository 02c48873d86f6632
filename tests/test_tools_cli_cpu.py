"""CLI smokes for the remaining tools/ entry points (train.py is covered
by test_train_cpu): generation.py, eval.py and export.py run end-to-end
on tiny configs exactly as a user would launch them."""

import os
import subprocess
import sys

import pytest

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))
GEN_CFG = os.path.join(REPO, "paddlefleetx_amd/configs/nlp/gpt",
                       "generation_gpt_345M_single_card.yaml")
EVAL_CFG = os.path.join(REPO, "paddlefleetx_amd/configs/nlp/gpt",
                        "pretrain_gpt_345M_single_card.yaml")

TINY_MODEL = ["Model.hidden_size=64", "Model.num_layers=2",
              "Model.num_attention_heads=4",
              "Model.max_position_embeddings=64", "Model.vocab_size=256",
              "Model.hidden_dropout_prob=0.0",
              "Model.attention_probs_dropout_prob=0.0"]


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                       cwd=REPO)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    return r.stdout + r.stderr


@pytest.mark.timeout(300)
def test_generation_cli():
    cmd = [sys.executable, os.path.join(REPO, "tools", "generation.py"),
           "-c", GEN_CFG, "--input-ids", "5,17,101"]
    for o in TINY_MODEL + ["Generation.max_dec_len=8",
                           "Generation.top_k=1",
                           "Generation.eos_token_id=255"]:
        cmd += ["-o", o]
    out = _run(cmd)
    assert "generated ids" in out


@pytest.mark.timeout(300)
def test_eval_cli():
    cmd = [sys.executable, os.path.join(REPO, "tools", "eval.py"),
           "-c", EVAL_CFG]
    for o in TINY_MODEL + [
            "Data.Eval.dataset.name=GPTSyntheticDataset",
            "Data.Eval.dataset.seq_len=64",
            "Data.Eval.dataset.vocab_size=256",
            "Data.Eval.dataset.num_samples=8",
            "Data.Eval.loader.num_workers=0",
            "Global.micro_batch_size=2", "Global.local_batch_size=2",
            "Global.global_batch_size=2",
            "Offline_Eval.max_iters=2",
            "Engine.mix_precision.enable=False"]:
        cmd += ["-o", o]
    out = _run(cmd)
    assert "eval" in out.lower()


@pytest.mark.timeout(300)
def test_export_cli(tmp_path):
    cmd = [sys.executable, os.path.join(REPO, "tools", "export.py"),
           "-c", EVAL_CFG, "--output-dir", str(tmp_path)]
    for o in TINY_MODEL + ["Engine.mix_precision.enable=False"]:
        cmd += ["-o", o]
    _run(cmd)
    files = os.listdir(tmp_path)
    assert any(f.endswith(".json") for f in files), files
    assert any("rank" in f or f.endswith((".safetensors", ".pdparams",
                                          ".pt")) for f in files), files


@pytest.mark.timeout(300)
def test_export_then_inference_cli(tmp_path):
    # export a tiny model, then run tools/inference.py over the artifact
    cmd = [sys.executable, os.path.join(REPO, "tools", "export.py"),
           "-c", GEN_CFG, "--output-dir", str(tmp_path)]
    for o in TINY_MODEL + ["Engine.mix_precision.enable=False",
                           "Generation.eos_token_id=255",
                           "Generation.max_dec_len=6"]:
        cmd += ["-o", o]
    _run(cmd)
    cmd = [sys.executable, os.path.join(REPO, "tools", "inference.py"),
           "-c", GEN_CFG, "--model-dir", str(tmp_path),
           "--input-ids", "5,17,101"]
    for o in ["Generation.eos_token_id=255", "Generation.max_dec_len=6"]:
        cmd += ["-o", o]
    out = _run(cmd)
    assert "generated ids" in out


@pytest.mark.timeout(300)
def test_auto_cli():
    cmd = [sys.executable, os.path.join(REPO, "tools", "auto.py"),
           "-c", EVAL_CFG]
    for o in TINY_MODEL + [
            "Data.Train.dataset.seq_len=64",
            "Data.Train.dataset.vocab_size=256",
            "Data.Train.dataset.num_samples=32",
            "Data.Train.loader.num_workers=0",
            "Global.micro_batch_size=2", "Global.local_batch_size=4",
            "Global.global_batch_size=4", "Global.max_steps=2",
            "Global.logging_freq=1", "Global.eval_freq=",
            "Global.save_steps=", "Engine.mix_precision.enable=False"]:
        cmd += ["-o", o]
    out = _run(cmd)
    assert "ips_total" in out


@pytest.mark.timeout(300)
def test_auto_export_cli(tmp_path):
    cmd = [sys.executable, os.path.join(REPO, "tools", "auto_export.py"),
           "-c", EVAL_CFG, "--output-dir", str(tmp_path)]
    for o in TINY_MODEL + ["Engine.mix_precision.enable=False"]:
        cmd += ["-o", o]
    _run(cmd)
    assert any(f.endswith(".json") for f in os.listdir(tmp_path))


@pytest.mark.timeout(300)
def test_inference_benchmark_cli(tmp_path):
    cmd = [sys.executable, os.path.join(REPO, "tools", "export.py"),
           "-c", GEN_CFG, "--output-dir", str(tmp_path)]
    for o in TINY_MODEL + ["Engine.mix_precision.enable=False",
                           "Generation.eos_token_id=255",
                           "Generation.max_dec_len=4"]:
        cmd += ["-o", o]
    _run(cmd)
    out = _run([sys.executable,
                os.path.join(REPO, "projects/gpt/benchmark.py"),
                "--model-dir", str(tmp_path), "--seq-len", "8",
                "--iter", "2"])
    assert "run time" in out and "ms/token" in out
