"""The decomposed-API example (examples/transformer — the reference's
newer engine-less training loop) stays runnable: 2 tiny CPU steps via
subprocess, exactly as a user would launch it."""

import os
import subprocess
import sys

import pytest

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


@pytest.mark.timeout(300)
def test_decomposed_pretrain_example_runs():
    script = os.path.join(REPO, "examples/transformer/models/GPT/pretrain",
                          "run.py")
    cmd = [sys.executable, script, "--max-steps", "2",
           "-o", "Model.hidden_size=64", "-o", "Model.num_layers=2",
           "-o", "Model.num_attention_heads=2", "-o", "Model.vocab_size=256",
           "-o", "Model.max_position_embeddings=64",
           "-o", "Model.hidden_dropout_prob=0.0",
           "-o", "Model.attention_probs_dropout_prob=0.0",
           "-o", "Global.local_batch_size=4",
           "-o", "Global.global_batch_size=4",
           "-o", "Global.micro_batch_size=4",
           "-o", "Data.Train.dataset.vocab_size=256",
           "-o", "Data.Train.dataset.seq_len=64",
           "-o", "Data.Train.loader.num_workers=0"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=240,
                       cwd=REPO)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "loss" in r.stdout + r.stderr


TINY = ["-o", "Model.hidden_size=64", "-o", "Model.num_layers=2",
        "-o", "Model.num_attention_heads=4", "-o", "Model.vocab_size=256",
        "-o", "Model.max_position_embeddings=64",
        "-o", "Model.hidden_dropout_prob=0.0",
        "-o", "Model.attention_probs_dropout_prob=0.0",
        "-o", "Engine.mix_precision.enable=False",
        "-o", "Global.micro_batch_size=2", "-o", "Global.local_batch_size=2"]


def _run_example(rel, extra):
    script = os.path.join(REPO, "examples/transformer/models/GPT", rel,
                          "run.py")
    r = subprocess.run([sys.executable, script] + TINY + extra,
                       capture_output=True, text=True, timeout=240, cwd=REPO)
    assert r.returncode == 0, r.stdout[-1500:] + r.stderr[-1500:]
    return r.stdout + r.stderr


@pytest.mark.timeout(300)
def test_decomposed_generation_example():
    out = _run_example("generation", ["-o", "Generation.eos_token_id=255",
                                      "-o", "Generation.max_dec_len=4"])
    assert "generated ids" in out


@pytest.mark.timeout(300)
def test_decomposed_offline_eval_example():
    out = _run_example("offline-eval", [
        "-o", "Data.Eval.dataset.vocab_size=256",
        "-o", "Data.Eval.dataset.seq_len=64",
        "-o", "Data.Eval.loader.num_workers=0", "--max-iters", "2"])
    assert "ppl" in out


@pytest.mark.timeout(300)
def test_decomposed_finetune_example():
    out = _run_example("finetune", [
        "-o", "Data.Train.dataset.vocab_size=256",
        "-o", "Data.Train.dataset.max_length=32",
        "-o", "Data.Eval.dataset.vocab_size=256",
        "-o", "Data.Eval.dataset.max_length=32",
        "-o", "Data.Train.loader.num_workers=0",
        "-o", "Data.Eval.loader.num_workers=0", "--max-steps", "2"])
    assert "finetune eval" in out


@pytest.mark.timeout(300)
def test_decomposed_moe_example():
    script = os.path.join(REPO, "examples/transformer/models/GPT",
                          "pretrain_moe", "run.py")
    r = subprocess.run([sys.executable, script,
                        "-o", "Model.hidden_size=64",
                        "-o", "Model.num_layers=2",
                        "-o", "Model.num_attention_heads=4",
                        "-o", "Model.vocab_size=256",
                        "-o", "Model.max_position_embeddings=64",
                        "-o", "Data.Train.dataset.vocab_size=256",
                        "-o", "Data.Train.dataset.seq_len=64",
                        "-o", "Data.Train.loader.num_workers=0",
                        "-o", "Global.micro_batch_size=2",
                        "-o", "Global.local_batch_size=2",
                        "-o", "Global.global_batch_size=2",
                        "-o", "Engine.mix_precision.enable=False",
                        "--max-steps", "2"],
                       capture_output=True, text=True, timeout=240, cwd=REPO)
    assert r.returncode == 0, r.stdout[-1500:] + r.stderr[-1500:]
    assert "loss" in r.stdout + r.stderr
