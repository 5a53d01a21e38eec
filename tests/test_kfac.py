"""K-FAC preconditioner tests (CPU).

Covers layer discovery/skipping, the preconditioning math against an
independent eigh computation, state-dict round-trip, end-to-end loss
descent on a tiny model, and factor averaging across a 2-rank gloo
group (reference dependency being replaced: kfac_pytorch, wired at
run_pretraining.py:321-355 of the reference).
"""

import math
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch import nn

from bert_pytorch_amd.config import BertConfig
from bert_pytorch_amd.models import BertForPreTraining, BertPretrainingCriterion
from bert_pytorch_amd.optim.kfac import KFAC


def _tiny_config():
    return BertConfig(
        vocab_size_or_config_json_file=128, hidden_size=32,
        num_hidden_layers=2, num_attention_heads=2,
        intermediate_size=64, max_position_embeddings=32,
    )


def _tiny_batch(gen=None):
    ids = torch.randint(0, 128, (4, 16), generator=gen)
    mask = torch.ones(4, 16, dtype=torch.long)
    labels = torch.full((4, 16), -1, dtype=torch.long)
    labels[:, 2:5] = torch.randint(0, 128, (4, 3), generator=gen)
    nsp = torch.zeros(4, dtype=torch.long)
    return ids, mask, labels, nsp


def test_layer_discovery_skips_lm_head_and_embeddings():
    model = BertForPreTraining(_tiny_config())
    kfac = KFAC(model)
    names = [st.name for st in kfac.layers]
    assert names, "no layers registered"
    for n in names:
        assert "embedding" not in n.lower()
        assert "cls.predictions" not in n  # BertLMPredictionHead subtree
    # attention QKV (packed), attention output, FFN output, pooler-adjacent
    assert any("attention" in n for n in names)


def test_precondition_matches_manual_eigh():
    torch.manual_seed(0)
    lin = nn.Linear(8, 6)
    model = nn.Sequential(lin)
    kfac = KFAC(model, damping=0.01, kl_clip=1e9)  # huge clip => nu == 1
    x = torch.randn(32, 8)
    y = model(x).sum()
    y.backward()
    kfac.step()

    st = kfac.layers[0]
    pi = math.sqrt(kfac.damping)
    for factor, inv in ((st.A, st.A_inv), (st.G, st.G_inv)):
        d, q = torch.linalg.eigh(factor + pi * torch.eye(factor.shape[0]))
        expect = (q / d.clamp_min(1e-10)) @ q.t()
        assert torch.allclose(inv, expect, rtol=1e-4, atol=1e-5)

    # reconstruct the expected preconditioned grad from a fresh backward
    lin2 = nn.Linear(8, 6)
    lin2.load_state_dict(lin.state_dict())
    model2 = nn.Sequential(lin2)
    model2(x).sum().backward()
    m = torch.cat([lin2.weight.grad, lin2.bias.grad.unsqueeze(1)], dim=1)
    v = st.G_inv @ m @ st.A_inv
    assert torch.allclose(lin.weight.grad, v[:, :-1], rtol=1e-4, atol=1e-5)
    assert torch.allclose(lin.bias.grad, v[:, -1], rtol=1e-4, atol=1e-5)


def test_kl_clip_scales_update():
    torch.manual_seed(0)
    lin = nn.Linear(4, 4)
    kfac = KFAC(nn.Sequential(lin), damping=0.01, kl_clip=1e-12, lr=1.0)
    x = torch.randn(16, 4)
    (lin(x) ** 2).sum().backward()
    raw = lin.weight.grad.clone()
    kfac.step()
    assert lin.weight.grad.abs().max() < raw.abs().max()


def test_end_to_end_loss_decreases():
    torch.manual_seed(0)
    config = _tiny_config()
    model = BertForPreTraining(config)
    criterion = BertPretrainingCriterion(config.vocab_size)
    opt = torch.optim.SGD(model.parameters(), lr=0.5)
    kfac = KFAC(model, optimizer=opt, inv_update_interval=2, damping=0.03)
    gen = torch.Generator().manual_seed(7)
    ids, mask, labels, nsp = _tiny_batch(gen)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        scores, rel, glabels = model(ids, None, mask, masked_lm_labels=labels)
        loss = criterion(scores, rel, glabels, nsp)
        loss.backward()
        kfac.step()
        opt.step()
        losses.append(float(loss))
        assert losses[-1] == losses[-1], "NaN loss"
    assert losses[-1] < losses[0]


def test_state_dict_roundtrip():
    torch.manual_seed(0)
    model = BertForPreTraining(_tiny_config())
    criterion = BertPretrainingCriterion(128)
    kfac = KFAC(model)
    ids, mask, labels, nsp = _tiny_batch()
    scores, rel, glabels = model(ids, None, mask, masked_lm_labels=labels)
    criterion(scores, rel, glabels, nsp).backward()
    kfac.step()
    state = kfac.state_dict()

    model2 = BertForPreTraining(_tiny_config())
    kfac2 = KFAC(model2)
    kfac2.load_state_dict(state)
    assert kfac2._steps == kfac._steps
    for a, b in zip(kfac.layers, kfac2.layers):
        if a.A is None:
            assert b.A is None
            continue
        assert torch.equal(a.A, b.A)
        assert torch.equal(a.G_inv, b.G_inv)


def _run_kfac_rank(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.manual_seed(0)  # identical weights on both ranks
        lin = nn.Linear(8, 6)
        model = nn.Sequential(lin)
        kfac = KFAC(model, damping=0.01, kl_clip=1e9)
        g = torch.Generator().manual_seed(50 + rank)  # different data
        x = torch.randn(16, 8, generator=g)
        model(x).sum().backward()
        dist.all_reduce(lin.weight.grad)
        lin.weight.grad /= world
        dist.all_reduce(lin.bias.grad)
        lin.bias.grad /= world
        kfac.step()
        # after step: factors all-reduced, inverses broadcast -> the
        # preconditioned grads must be bit-identical across ranks
        flat = torch.cat([lin.weight.grad.reshape(-1), lin.bias.grad])
        peer = flat.clone()
        dist.broadcast(peer, src=0)
        q.put((rank, bool(torch.equal(flat, peer))))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e}"))


@pytest.mark.timeout(120)
def test_kfac_distributed_gloo():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_run_kfac_rank, args=(r, 2, 29533, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=110) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"
