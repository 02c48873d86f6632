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
