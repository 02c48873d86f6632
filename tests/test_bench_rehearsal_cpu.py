"""Rehearse the driver's bench.py invocation shape on CPU/gloo so the
first 8-GPU driver run is not the first execution of those code paths
(VERDICT r1 missing #1): launches bench.py through
`python -m torch.distributed.run --nnodes=1 --nproc-per-node N
--master-addr 127.0.0.1` exactly as the driver does, at N=2 (DP2) and
N=8 (DP2xTP2xPP2), with the GPT-TEST model, and checks the single JSON
line contract."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.join(os.path.dirname(__file__), "..")


def _run_bench(n):
    from port_util import free_port
    port = free_port()
    env = dict(os.environ)
    env.pop("RANK", None); env.pop("WORLD_SIZE", None)
    env.pop("MASTER_ADDR", None); env.pop("MASTER_PORT", None)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), "bench.py", "--gpus", str(n),
           "--steps", "2", "--warmup", "1", "--model", "GPT-TEST",
           "--seq-len", "32", "--micro-batch", "2", "--acc-steps", "2"]
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                         text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-3000:]
    json_lines = [l for l in out.stdout.splitlines()
                  if l.startswith("{") and '"metric"' in l]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == n
    assert rec["steps"] == 2
    assert rec["value"] > 0
    assert rec["unit"] == "tokens/s"
    return rec


@pytest.mark.timeout(600)
def test_bench_world2_dp2():
    rec = _run_bench(2)
    assert rec["config"]["parallelism"] == "dp2tp1pp1"


@pytest.mark.timeout(600)
def test_bench_world8_dp2tp2pp2():
    rec = _run_bench(8)
    assert rec["config"]["parallelism"] == "dp2tp2pp2"
    assert rec["config"]["loss"] is not None


def test_bench_defaults_contract():
    """Driver-contract guard: model table, topology mapping and the
    measured micro/acc defaults must not drift."""
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "bench", os.path.join(REPO, "bench.py"))
    bench = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(bench)
    assert bench.topology_for(1) == (1, 1, 1)
    assert bench.topology_for(2) == (2, 1, 1)
    assert bench.topology_for(4) == (4, 1, 1)
    assert bench.topology_for(8) == (2, 2, 2)
    assert bench.MODELS["GPT-6.7B"] == dict(hidden_size=4096, num_layers=32,
                                            num_attention_heads=32)
    assert bench.MODELS["GPT-13B"] == dict(hidden_size=5120, num_layers=40,
                                           num_attention_heads=40)
    src = open(os.path.join(REPO, "bench.py")).read()
    # measured defaults (profiles/r02_micro16_ab.txt): micro16/acc2 at
    # pp==1, micro4/acc8 under pipeline, 13B capped at micro4
    assert "args.micro_batch or 16" in src
    assert "args.acc_steps or 2" in src
    assert "args.micro_batch or 4" in src
