"""Optimizer and scheduler behavior tests (eager paths; GPU parity of the
HIP kernels against these same implementations is in test_gpu_kernels)."""

import math

import pytest
import torch

from bert_pytorch_amd.optim import (
    BertAdam,
    ConstantWarmUpScheduler,
    CosineWarmUpScheduler,
    FusedAdam,
    FusedLAMB,
    LinearWarmUpScheduler,
    PolyWarmUpScheduler,
    clip_grad_norm_,
    warmup_exp_decay_exp,
)


def _quadratic_problem(seed=0, n=4):
    g = torch.Generator().manual_seed(seed)
    params = [torch.randn(8, 8, generator=g, requires_grad=True),
              torch.randn(16, generator=g, requires_grad=True)]
    return params


def _loss(params):
    return sum((p**2).sum() for p in params)


def test_lamb_decreases_loss():
    params = _quadratic_problem()
    opt = FusedLAMB(params, lr=0.05)
    initial = float(_loss(params))
    for _ in range(50):
        opt.zero_grad()
        loss = _loss(params)
        loss.backward()
        opt.step()
    assert float(_loss(params)) < initial * 0.5


def test_lamb_tracks_group_step():
    params = _quadratic_problem()
    opt = FusedLAMB(params, lr=0.01)
    assert "step" not in opt.param_groups[0]
    loss = _loss(params)
    loss.backward()
    opt.step()
    assert opt.param_groups[0]["step"] == 1
    opt.step()
    assert opt.param_groups[0]["step"] == 2


def test_lamb_trust_ratio_only_on_decay_group():
    """wd=0 group takes a plain Adam step (use_nvlamb=False)."""
    torch.manual_seed(0)
    p_decay = torch.full((4,), 2.0, requires_grad=True)
    p_plain = torch.full((4,), 2.0, requires_grad=True)
    opt = FusedLAMB(
        [
            {"params": [p_decay], "weight_decay": 0.01},
            {"params": [p_plain], "weight_decay": 0.0},
        ],
        lr=0.1, max_grad_norm=1e9,
    )
    p_decay.grad = torch.full((4,), 0.5)
    p_plain.grad = torch.full((4,), 0.5)
    opt.step()
    # plain group: update magnitude = lr * mhat/(sqrt(vhat)+eps) ~= lr
    assert abs((2.0 - float(p_plain[0])) - 0.1) < 1e-3
    # decay group scaled by trust ratio ||w||/||u|| != 1
    assert abs(2.0 - float(p_decay[0])) > 0.15


def test_lamb_global_clip():
    p = torch.zeros(10, requires_grad=True)
    opt = FusedLAMB([p], lr=0.1, max_grad_norm=1.0, weight_decay=0.0)
    p.grad = torch.full((10,), 100.0)
    opt.step()  # huge grad clipped to norm 1; step still sane
    assert float(p.abs().max()) < 0.2


def test_adam_matches_torch_adamw():
    torch.manual_seed(1)
    p1 = torch.randn(32, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    mine = FusedAdam([p1], lr=1e-2, weight_decay=0.01, eps=1e-8)
    ref = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.01, eps=1e-8)
    for i in range(10):
        g = torch.randn(32, generator=torch.Generator().manual_seed(i))
        p1.grad = g.clone()
        p2.grad = g.clone()
        mine.step()
        ref.step()
    # AdamW couples wd with p BEFORE update; ours adds wd*p to update --
    # equivalent to first order, close numerically at small lr
    torch.testing.assert_close(p1, p2, rtol=2e-3, atol=2e-3)


def test_bert_adam_schedule_and_step():
    p = torch.randn(16, requires_grad=True)
    opt = BertAdam([p], lr=1e-3, warmup=0.1, t_total=100)
    for _ in range(5):
        opt.zero_grad()
        (p**2).sum().backward()
        opt.step()
    assert opt.get_lr()[0] > 0


def test_poly_scheduler_follows_optimizer_step():
    params = _quadratic_problem()
    opt = FusedLAMB(params, lr=1.0)
    sched = PolyWarmUpScheduler(opt, warmup=0.1, total_steps=100)
    lrs = []
    for step in range(20):
        opt.zero_grad()
        _loss(params).backward()
        sched.step()
        opt.step()
        lrs.append(opt.param_groups[0]["lr"])
    # warmup: lr rises for first ~10 steps then decays
    assert lrs[4] < lrs[8]
    assert lrs[15] > lrs[19]


def test_poly_scheduler_phase2_resume():
    """Scheduler picks up from the optimizer's step after state surgery."""
    params = _quadratic_problem()
    opt = FusedLAMB(params, lr=1.0)
    for group in opt.param_groups:
        group["step"] = 50
    sched = PolyWarmUpScheduler(opt, warmup=0.1, total_steps=100)
    sched.step()
    expected = 1.0 * (1.0 - 51 / 100) ** 0.5
    assert math.isclose(opt.param_groups[0]["lr"], expected, rel_tol=1e-6)


def test_linear_constant_schedulers():
    params = _quadratic_problem()
    opt = FusedAdam(params, lr=1.0)
    s = LinearWarmUpScheduler(opt, warmup=0.5, total_steps=10)
    opt.param_groups[0]["step"] = 9
    s.step()
    assert opt.param_groups[0]["lr"] == 0.0
    opt2 = FusedAdam(_quadratic_problem(), lr=1.0)
    c = ConstantWarmUpScheduler(opt2, warmup=0.2, total_steps=10)
    for _ in range(5):
        c.step()
    assert opt2.param_groups[0]["lr"] == 1.0


def test_cosine_scheduler_matches_reference_formula():
    """Cosine decay reproduces the reference's exact (quirky) formula
    lr * 0.5 * (1 + cos(pi + progress)) (src/schedulers.py:61-66) --
    behavior parity matters more than the textbook curve."""
    import math

    opt = FusedAdam(_quadratic_problem(), lr=1.0)
    sch = CosineWarmUpScheduler(opt, warmup=0.2, total_steps=10)
    sch.step()  # last_epoch 0 -> 1: progress 0.1, in warmup
    assert opt.param_groups[0]["lr"] == pytest.approx(0.1 / 0.2)
    for _ in range(4):
        sch.step()  # last_epoch -> 5: progress 0.5, past warmup
    expected = 0.5 * (1.0 + math.cos(math.pi + 0.5))
    assert opt.param_groups[0]["lr"] == pytest.approx(expected)


def test_warmup_exp_decay_exp():
    assert warmup_exp_decay_exp(0, 0.9, 100, 1000, warmup=0.0) == 1.0
    mid = warmup_exp_decay_exp(5, 0.9, 100, 1000, warmup=0.01)
    assert 0 < mid <= 1.0


def test_clip_grad_norm():
    p = torch.zeros(100, requires_grad=True)
    p.grad = torch.full((100,), 1.0)
    norm = clip_grad_norm_([p], max_norm=1.0)
    assert math.isclose(norm, 10.0, rel_tol=1e-5)
    assert math.isclose(float(p.grad.norm()), 1.0, rel_tol=1e-5)


def test_optimizer_state_dict_roundtrip():
    params = _quadratic_problem()
    opt = FusedLAMB(params, lr=0.01)
    for _ in range(3):
        opt.zero_grad()
        _loss(params).backward()
        opt.step()
    state = opt.state_dict()
    params2 = _quadratic_problem()
    opt2 = FusedLAMB(params2, lr=0.01)
    opt2.load_state_dict(state)
    # the group 'step' key survives (required by Poly/Linear schedulers)
    assert opt2.param_groups[0]["step"] == 3


def test_fused_lamb_master_weights_bf16():
    """bf16 params + fp32 masters: loss decreases, master holds the
    precise weights, params are the cast-down of the master."""
    import torch
    from bert_pytorch_amd.optim import FusedLAMB

    torch.manual_seed(0)
    lin = torch.nn.Linear(16, 16).to(torch.bfloat16)
    opt = FusedLAMB(lin.parameters(), lr=1e-2, master_weights=True)
    x = torch.randn(32, 16).to(torch.bfloat16)
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = (lin(x) ** 2).float().mean()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0]
    state = opt.state[lin.weight]
    assert state["master"].dtype == torch.float32
    assert torch.equal(lin.weight.data,
                       state["master"].to(torch.bfloat16))
    # state_dict round-trip preserves the master copy
    sd = opt.state_dict()
    lin2 = torch.nn.Linear(16, 16).to(torch.bfloat16)
    opt2 = FusedLAMB(lin2.parameters(), lr=1e-2, master_weights=True)
    opt2.load_state_dict(sd)
    st2 = next(iter(opt2.state.values()))
    assert "master" in st2 and st2["master"].dtype == torch.float32
