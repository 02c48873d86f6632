import os
import textwrap

import pytest

from paddlefleetx_amd.utils.config import get_config

CFG_DIR = os.path.join(os.path.dirname(__file__), "..",
                       "paddlefleetx_amd", "configs", "nlp", "gpt")


def test_base_inheritance_and_derivation():
    cfg = get_config(os.path.join(CFG_DIR, "pretrain_gpt_345M_single_card.yaml"))
    assert cfg.Model.hidden_size == 1024
    assert cfg.Global.global_batch_size == 8
    assert cfg.Engine.accumulate_steps == 1
    assert cfg.Model.ffn_hidden_size == 4096  # derived 4*h
    assert cfg.Distributed.dp_degree == 1
    assert cfg.Model.padded_vocab_size == 50304


def test_cli_overrides():
    cfg = get_config(os.path.join(CFG_DIR, "pretrain_gpt_345M_single_card.yaml"),
                     overrides=["Model.num_layers=2", "Optimizer.lr.max_lr=1e-3",
                                "Global.micro_batch_size=4"])
    assert cfg.Model.num_layers == 2
    assert cfg.Optimizer.lr.max_lr == pytest.approx(1e-3)
    assert cfg.Engine.accumulate_steps == 2  # 8 local / 4 micro


def test_batch_math_validation(tmp_path):
    p = tmp_path / "bad.yaml"
    p.write_text(textwrap.dedent("""
        Global:
          global_batch_size: 7
        Model:
          name: GPTModule
        Distributed:
          dp_degree: 2
          world_size: 2
    """))
    with pytest.raises(AssertionError):
        get_config(str(p))


def test_nested_base(tmp_path):
    (tmp_path / "a.yaml").write_text("Global:\n  local_batch_size: 4\n  seed: 7\n")
    (tmp_path / "b.yaml").write_text("_base_: ./a.yaml\nGlobal:\n  seed: 9\n")
    cfg = get_config(str(tmp_path / "b.yaml"))
    assert cfg.Global.seed == 9
    assert cfg.Global.local_batch_size == 4
