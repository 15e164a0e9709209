"""FusedAdam (flat master buffers) vs torch.optim.Adam parity on CPU."""
import os

import torch
import torch.multiprocessing as mp

from pertgnn.models import SAGEDeterministic
from pertgnn.train.optim import FusedAdam


def _model(seed=0):
    torch.manual_seed(seed)
    return SAGEDeterministic(9, [5], 3, 3, 2, hidden_channels=8, num_layers=1, dropout=0.0)


def _inputs(seed=1, n=20, e=40, b=3):
    g = torch.Generator().manual_seed(seed)
    return dict(
        x=torch.randn(n, 9, generator=g),
        cat_X=torch.randint(0, 5, (n, 1), generator=g),
        edge_index=torch.stack([torch.randint(0, n, (e,), generator=g),
                                torch.randint(0, n, (e,), generator=g)]),
        edge_attr=torch.stack([torch.randint(0, 4, (e,), generator=g),
                               torch.randint(0, 3, (e,), generator=g)], dim=1),
        pattern_num_nodes=torch.randint(1, 4, (n, 1), generator=g).float(),
        pattern_probs=torch.rand(n, 1, generator=g),
        entry_id=torch.randint(0, 4, (b,), generator=g),
        batch=torch.sort(torch.randint(0, b, (n,), generator=g)).values,
    ), torch.rand(b, generator=g) * 5


def _train(model, opt, steps=5):
    from pertgnn.ops import functional as F
    inp, y = _inputs()
    b = int(inp["batch"].max()) + 1
    for _ in range(steps):
        opt.zero_grad()
        gp, _ = model(**inp)
        loss = F.quantile_loss(y[:b], gp.flatten(), 0.5)
        loss.backward()
        opt.step()
    return {n: p.detach().clone() for n, p in model.named_parameters()
            if not isinstance(p, torch.nn.parameter.UninitializedParameter)}


def test_fused_adam_matches_torch_adam_one_step():
    """Exact-formula check: after ONE step params agree to fp32 roundoff.
    (Multi-step trajectories diverge chaotically from 1-ulp differences when
    early-step grads are near zero — the GPU test covers multi-step parity on
    well-conditioned grads.)"""
    m1 = _model()
    p1 = _train(m1, torch.optim.Adam(m1.parameters(), lr=1e-2), steps=1)
    m2 = _model()
    p2 = _train(m2, FusedAdam(m2.parameters(), lr=1e-2), steps=1)
    for n in p1:
        assert torch.allclose(p1[n], p2[n], atol=1e-6), (n, (p1[n] - p2[n]).abs().max())


def test_fused_adam_multi_step_close():
    m1 = _model()
    p1 = _train(m1, torch.optim.Adam(m1.parameters(), lr=1e-2))
    m2 = _model()
    p2 = _train(m2, FusedAdam(m2.parameters(), lr=1e-2))
    for n in p1:
        assert torch.allclose(p1[n], p2[n], atol=5e-3), (n, (p1[n] - p2[n]).abs().max())


def test_fused_adam_param_views_preserved():
    m = _model()
    before = {n: p.detach().clone() for n, p in m.named_parameters()
              if not isinstance(p, torch.nn.parameter.UninitializedParameter)}
    opt = FusedAdam(m.parameters(), lr=1e-3)
    for n, p in m.named_parameters():
        if isinstance(p, torch.nn.parameter.UninitializedParameter):
            continue
        assert torch.allclose(p.detach(), before[n])
        # storage shared with the flat buffer
        assert p.data_ptr() >= opt.flat_param.data_ptr()


def test_fused_adam_state_roundtrip(tmp_path):
    m = _model()
    opt = FusedAdam(m.parameters(), lr=1e-2)
    _train_steps = _train(m, opt, steps=3)
    sd = {k: (v.clone() if torch.is_tensor(v) else v) for k, v in opt.state_dict().items()}
    m2 = _model()
    opt2 = FusedAdam(m2.parameters(), lr=1e-2)
    opt2.load_state_dict(sd)
    assert opt2.step_count == opt.step_count
    assert torch.allclose(opt2.exp_avg, opt.exp_avg)


def test_fused_adam_loss_scaling_equivalence():
    """grad_scale=S with S-scaled gradients reproduces the unscaled step."""
    from pertgnn.train.optim import FusedAdam

    torch.manual_seed(5)
    w0 = torch.randn(64, 8)
    g = torch.randn(64, 8) * 1e-3

    p_a = torch.nn.Parameter(w0.clone())
    opt_a = FusedAdam([p_a], lr=1e-3)
    p_a.grad.copy_(g)
    opt_a.step()

    S = 1024.0
    p_b = torch.nn.Parameter(w0.clone())
    opt_b = FusedAdam([p_b], lr=1e-3, grad_scale=S)
    p_b.grad.copy_(g * S)
    opt_b.step()

    assert torch.allclose(p_a.data, p_b.data, atol=1e-7), \
        (p_a.data - p_b.data).abs().max()


def test_dynamic_loss_scaling_cpu():
    """Dynamic scaler semantics (eager path): overflow steps are skipped
    with backoff, clean steps match the unscaled trajectory, and the scale
    grows after growth_interval clean steps."""
    from pertgnn.train.optim import FusedAdam

    torch.manual_seed(0)
    m = torch.nn.Linear(8, 4)
    opt = FusedAdam(m.parameters(), lr=1e-2, dynamic_scale=True,
                    init_scale=8.0, growth_interval=3)
    x = torch.randn(16, 8)
    for i in range(8):
        opt.zero_grad()
        loss = m(x).pow(2).mean()
        opt.scale_loss(loss).backward()
        if i == 2:
            opt.flat_grad[0] = float("inf")
        before = opt.flat_param.clone()
        sc_before = float(opt.sstate[0])
        opt.step()
        if i == 2:
            assert torch.equal(before, opt.flat_param), "overflow must skip"
            assert float(opt.sstate[0]) == sc_before * 0.5
        else:
            assert not torch.equal(before, opt.flat_param)

    torch.manual_seed(0)
    m2 = torch.nn.Linear(8, 4)
    opt2 = FusedAdam(m2.parameters(), lr=1e-2)
    for i in range(8):
        if i == 2:
            continue
        opt2.zero_grad()
        m2(x).pow(2).mean().backward()
        opt2.step()
    assert torch.allclose(opt.flat_param, opt2.flat_param, atol=1e-6)
    assert opt.sstate is not None and "sstate" in opt.state_dict()
