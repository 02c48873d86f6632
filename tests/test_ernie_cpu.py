"""ERNIE: model forward/backward, heads, masking, module, dataset."""

import numpy as np
import pytest
import torch

from paddlefleetx_amd.parallel.env import set_hcg
from paddlefleetx_amd.parallel.topology import HybridTopology


@pytest.fixture(autouse=True)
def _env():
    set_hcg(HybridTopology())
    yield


def tiny_ernie(**kw):
    from paddlefleetx_amd.models.ernie import ErnieModel
    torch.manual_seed(0)
    cfg = dict(vocab_size=128, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=64, hidden_dropout_prob=0.0,
               attention_probs_dropout_prob=0.0)
    cfg.update(kw)
    return ErnieModel(**cfg)


def test_ernie_model_shapes():
    m = tiny_ernie()
    ids = torch.randint(0, 128, (2, 16))
    seq, pooled = m(ids)
    assert seq.shape == (2, 16, 64)
    assert pooled.shape == (2, 64)


def test_ernie_attention_mask_blocks_padding():
    m = tiny_ernie().eval()
    ids = torch.randint(4, 128, (1, 8))
    mask = torch.ones(1, 8)
    seq_full, _ = m(ids, attention_mask=mask)
    # padding the tail and masking it must not change the first tokens
    ids_pad = torch.cat([ids, torch.zeros(1, 4, dtype=torch.long)], dim=1)
    mask_pad = torch.cat([mask, torch.zeros(1, 4)], dim=1)
    seq_pad, _ = m(ids_pad, attention_mask=mask_pad)
    assert torch.allclose(seq_full[0, :8], seq_pad[0, :8], atol=1e-4)


def test_ernie_pretraining_loss_and_tied_weights():
    from paddlefleetx_amd.models.ernie import (ErnieForPretraining,
                                               ErniePretrainingCriterion)
    m = ErnieForPretraining(tiny_ernie())
    # decoder weight is the embedding weight (tied)
    assert m.cls.predictions.decoder_weight is \
        m.ernie.embeddings.word_embeddings.weight
    ids = torch.randint(0, 128, (2, 16))
    tt = torch.zeros_like(ids)
    pred, seq_rel = m(ids, tt)
    assert pred.shape == (2, 16, 128) and seq_rel.shape == (2, 2)
    labels = torch.randint(0, 128, (2, 16))
    labels[:, ::2] = -1
    nsp_labels = torch.randint(0, 2, (2,))
    crit = ErniePretrainingCriterion()
    mlm, nsp = crit(pred, seq_rel, labels, nsp_labels)
    (mlm + nsp).backward()
    assert m.ernie.embeddings.word_embeddings.weight.grad is not None


def test_ernie_masked_positions_gather():
    from paddlefleetx_amd.models.ernie import ErnieForPretraining
    m = ErnieForPretraining(tiny_ernie()).eval()
    ids = torch.randint(0, 128, (2, 8))
    pred_full, _ = m(ids)
    pos = torch.tensor([1, 5, 9])  # flattened positions into [2*8]
    pred_sel, _ = m(ids, masked_positions=pos)
    assert pred_sel.shape == (3, 128)
    flat = pred_full.reshape(-1, 128)
    assert torch.allclose(pred_sel, flat[pos], atol=1e-5)


def test_ernie_seq_cls():
    from paddlefleetx_amd.models.ernie import ErnieForSequenceClassification
    m = ErnieForSequenceClassification(tiny_ernie(), num_classes=3)
    logits = m(torch.randint(0, 128, (4, 10)))
    assert logits.shape == (4, 3)


def test_masking_statistics():
    from paddlefleetx_amd.data.ernie_dataset import \
        create_masked_lm_predictions
    rng = np.random.RandomState(0)
    tokens = rng.randint(4, 1000, size=512).astype(np.int64)
    masked, labels = create_masked_lm_predictions(tokens, 1000, rng,
                                                  masked_lm_prob=0.15)
    n_pred = (labels >= 0).sum()
    assert 50 <= n_pred <= 100  # ~15% of 512
    # labels hold originals where predicted
    sel = labels >= 0
    assert (labels[sel] == tokens[sel]).all()
    # most predicted positions became [MASK]=3
    frac_mask = (masked[sel] == 3).mean()
    assert 0.6 < frac_mask < 0.95


def test_ernie_synthetic_dataset_and_module():
    from paddlefleetx_amd.data.ernie_dataset import ErnieSyntheticDataset
    ds = ErnieSyntheticDataset(num_samples=4, seq_len=64, vocab_size=500)
    ids, tt, labels, nsp = ds[0]
    assert ids.shape == (64,) and tt.shape == (64,)
    assert nsp.item() in (0, 1)
    ids2, *_ = ds[0]
    assert torch.equal(ids, ids2)  # deterministic

    from paddlefleetx_amd.models import build_module
    cfg = {
        "Global": {"global_batch_size": 2},
        "Engine": {"mix_precision": {"enable": False}},
        "Model": {"name": "ErnieModule", "vocab_size": 500,
                  "hidden_size": 32, "num_hidden_layers": 1,
                  "num_attention_heads": 2, "intermediate_size": 64,
                  "max_position_embeddings": 64,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0},
    }
    mod = build_module(cfg)
    batch = ErnieSyntheticDataset.collate_fn([ds[0], ds[1]])
    loss = mod.training_step(batch)
    loss.backward()
    assert loss.ndim == 0 and float(loss) > 0


def test_ernie_moe_integration():
    m = tiny_ernie(moe_configs={"expert_mode": True, "num_experts": 4,
                                "gate": "naive", "top_k": 2})
    from paddlefleetx_amd.models.moe import MoELayer
    assert any(isinstance(mod, MoELayer) for mod in m.modules())
    seq, pooled = m(torch.randint(0, 128, (2, 8)))
    assert seq.shape == (2, 8, 64)


def test_ernie_finetune_config_end_to_end():
    """The shipped finetune_ernie yaml drives ErnieSeqClsModule + the
    seq-cls synthetic dataset through the engine."""
    import os
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.data import build_dataloader
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.utils.config import get_config
    repo = os.path.join(os.path.dirname(__file__), "..")
    cfg = get_config(os.path.join(
        repo, "paddlefleetx_amd/configs/nlp/ernie/"
              "finetune_ernie_345M_single_card.yaml"),
        overrides=["Model.hidden_size=32", "Model.num_hidden_layers=2",
                   "Model.num_attention_heads=2",
                   "Model.intermediate_size=64", "Model.vocab_size=200",
                   "Model.max_position_embeddings=64",
                   "Data.Train.dataset.seq_len=32",
                   "Data.Train.dataset.vocab_size=200",
                   "Data.Train.dataset.num_samples=16",
                   "Data.Train.loader.num_workers=0",
                   "Global.micro_batch_size=4", "Global.local_batch_size=4",
                   "Global.global_batch_size=4",
                   "Engine.mix_precision.enable=False"])
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    loader = build_dataloader(cfg, "Train")
    batch = next(iter(loader))
    assert len(batch) == 3 and batch[2].dtype == torch.long
    loss = engine._fit_impl(batch)
    assert torch.isfinite(loss)
