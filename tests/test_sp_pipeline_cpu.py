"""SP + pipeline composition: GPT pp2 x mp2 with sequence_parallel must
match the same pipe model without SP (identical TP shards), gloo world 4.

Reference allows Megatron-SP inside the pipeline (sequence_parallel_utils
used by GPTForPretrainingPipe's layers; partial-send-recv knob disabled
under SP, config.py:112-119).
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.join(os.path.dirname(__file__), "..")


def _worker(rank, world, port, tmpdir):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.models.gpt.model import GPTPretrainingCriterion
    from paddlefleetx_amd.models.gpt.pipeline_model import \
        GPTForPretrainingPipe
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(mp=2, pp=2)
    set_hcg(hcg)

    def build(sp):
        set_seed(1234)  # identical TP shard init across the two variants
        torch.manual_seed(9 + hcg.get_pipe_parallel_rank())
        return GPTForPretrainingPipe(
            vocab_size=128, hidden_size=32, num_layers=4,
            num_attention_heads=4, max_position_embeddings=32,
            hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
            fused_attn=False, sequence_parallel=sp, dtype=torch.float32)

    m_plain = build(False)
    m_sp = build(True)
    m_sp.load_state_dict(m_plain.state_dict())

    torch.manual_seed(5)
    batch = (torch.randint(0, 128, (4, 32)),
             torch.arange(32).unsqueeze(0).repeat(4, 1),
             torch.randint(0, 128, (4, 32)), torch.ones(4, 32))
    crit = GPTPretrainingCriterion()
    l_plain = m_plain.forward_backward_pipeline(batch, crit,
                                                accumulate_steps=2)
    l_sp = m_sp.forward_backward_pipeline(batch, crit, accumulate_steps=2)
    assert torch.allclose(l_plain, l_sp, atol=1e-5), (l_plain, l_sp)
    # grads must agree shard-for-shard (before the engine's SP-param
    # allreduce, compare only non-LN params; LN grads are per-shard
    # partial sums under SP)
    for (n1, p1), (n2, p2) in zip(m_plain.named_parameters(),
                                  m_sp.named_parameters()):
        assert n1 == n2
        if p1.grad is None or "ln" in n1 or "final" in n1:
            continue
        if getattr(p2, "sequence_parallel", False):
            continue
        assert torch.allclose(p1.grad, p2.grad, atol=1e-4), \
            (n1, (p1.grad - p2.grad).abs().max())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_gpt_sp_pipeline_matches_plain():
    from port_util import free_port; port = free_port()
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as tmpdir:
        procs = [ctx.Process(target=_worker, args=(r, 4, port, tmpdir))
                 for r in range(4)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _psr_worker(rank, world, port, tmpdir):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.models.gpt.model import GPTPretrainingCriterion
    from paddlefleetx_amd.models.gpt.pipeline_model import \
        GPTForPretrainingPipe
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(mp=2, pp=2)
    set_hcg(hcg)

    def build(psr):
        set_seed(1234)
        torch.manual_seed(9 + hcg.get_pipe_parallel_rank())
        return GPTForPretrainingPipe(
            vocab_size=128, hidden_size=32, num_layers=4,
            num_attention_heads=4, max_position_embeddings=32,
            hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
            fused_attn=False, partial_send_recv=psr, dtype=torch.float32)

    m_full = build(False)
    m_psr = build(True)
    m_psr.load_state_dict(m_full.state_dict())
    assert m_psr.partial_send_recv
    torch.manual_seed(5)
    batch = (torch.randint(0, 128, (4, 32)),
             torch.arange(32).unsqueeze(0).repeat(4, 1),
             torch.randint(0, 128, (4, 32)), torch.ones(4, 32))
    crit = GPTPretrainingCriterion()
    l_full = m_full.forward_backward_pipeline(batch, crit,
                                              accumulate_steps=2)
    l_psr = m_psr.forward_backward_pipeline(batch, crit,
                                            accumulate_steps=2)
    assert torch.allclose(l_full, l_psr, atol=1e-5), (l_full, l_psr)
    for (n1, p1), (n2, p2) in zip(m_full.named_parameters(),
                                  m_psr.named_parameters()):
        if p1.grad is None:
            continue
        assert torch.allclose(p1.grad, p2.grad, atol=1e-5), \
            (n1, (p1.grad - p2.grad).abs().max())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_partial_send_recv_matches_full():
    """enable_partial_send_recv: 1/mp chunked pp p2p + mp allgather must
    be exactly equivalent to full-tensor p2p (reference env.py:143)."""
    from port_util import free_port; port = free_port()
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as tmpdir:
        procs = [ctx.Process(target=_psr_worker, args=(r, 4, port, tmpdir))
                 for r in range(4)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"
