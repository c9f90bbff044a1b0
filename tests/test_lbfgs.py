"""LBFGSNew tests on deterministic convex/nonconvex problems
(line search must satisfy descent, history must be SPD-safe)."""

import math

import pytest
import torch

from fedkit.optim import LBFGSNew


def _run(opt, closure, steps):
    losses = []
    for _ in range(steps):
        loss = opt.step(closure)
        losses.append(float(loss.detach() if hasattr(loss, 'detach') else loss))
    return losses


def test_quadratic_full_batch_converges():
    torch.manual_seed(0)
    A = torch.randn(6, 6)
    A = A @ A.t() + 6 * torch.eye(6)       # SPD
    b = torch.randn(6)
    x = torch.zeros(6, requires_grad=True)

    opt = LBFGSNew([x], history_size=7, max_iter=10, line_search_fn=True,
                   batch_mode=False)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = 0.5 * x @ A @ x - b @ x
        if loss.requires_grad:
            loss.backward()
        return loss

    losses = _run(opt, closure, 6)
    x_star = torch.linalg.solve(A, b)
    f_star = float(0.5 * x_star @ A @ x_star - b @ x_star)
    assert losses[-1] < f_star + 1e-3
    assert losses[-1] <= losses[0]


def test_rosenbrock_decreases():
    x = torch.tensor([-1.2, 1.0], requires_grad=True)
    opt = LBFGSNew([x], history_size=10, max_iter=10, line_search_fn=True,
                   batch_mode=False)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = 100 * (x[1] - x[0] ** 2) ** 2 + (1 - x[0]) ** 2
        if loss.requires_grad:
            loss.backward()
        return loss

    losses = _run(opt, closure, 20)
    assert losses[-1] < 1.0          # from 24.2 at start
    assert not math.isnan(losses[-1])


def test_batch_mode_stochastic_quadratic():
    """Batch mode with changing minibatches must still reduce the full loss."""
    torch.manual_seed(3)
    n, d = 256, 10
    X = torch.randn(n, d)
    w_true = torch.randn(d)
    y = X @ w_true + 0.01 * torch.randn(n)
    w = torch.zeros(d, requires_grad=True)
    opt = LBFGSNew([w], history_size=7, max_iter=4, line_search_fn=True,
                   batch_mode=True)

    def full_loss():
        return float(((X @ w.detach() - y) ** 2).mean())

    start = full_loss()
    for it in range(12):
        sel = torch.randint(0, n, (64,))

        def closure():
            opt.zero_grad()
            loss = ((X[sel] @ w - y[sel]) ** 2).mean()
            if loss.requires_grad:
                loss.backward()
            return loss

        opt.step(closure)
    assert full_loss() < 0.25 * start


def test_history_gate_rejects_negative_curvature():
    """ys <= 1e-10||s||^2 pairs must not enter the history (lbfgsnew.py:618)."""
    x = torch.tensor([1.0], requires_grad=True)
    opt = LBFGSNew([x], history_size=5, max_iter=3, line_search_fn=False, lr=0.5)

    def closure():
        opt.zero_grad()
        loss = -(x ** 2).sum()       # concave: curvature always negative
        loss.backward()
        return loss

    opt.step(closure)
    state = opt.state[opt._params[0]]
    for yv, sv in zip(state["old_dirs"], state["old_stps"]):
        assert float(yv.dot(sv)) > 0


def test_zero_grad_early_exit():
    x = torch.tensor([0.0], requires_grad=True)
    opt = LBFGSNew([x], max_iter=5, line_search_fn=True, batch_mode=False)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = (x ** 2).sum()
        if loss.requires_grad:
            loss.backward()
        return loss

    loss = opt.step(closure)     # grad is 0 at the optimum: returns at once
    assert float(loss.detach()) == 0.0
    assert float(x.detach()) == 0.0


def test_multi_tensor_params():
    """Flat plumbing across several parameter tensors of different shapes."""
    torch.manual_seed(0)
    a = torch.randn(3, 2, requires_grad=True)
    b = torch.randn(4, requires_grad=True)
    target_a = torch.ones(3, 2)
    target_b = -torch.ones(4)
    opt = LBFGSNew([a, b], history_size=5, max_iter=10, line_search_fn=True)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = ((a - target_a) ** 2).sum() + ((b - target_b) ** 2).sum()
        if loss.requires_grad:
            loss.backward()
        return loss

    _run(opt, closure, 5)
    assert torch.allclose(a.detach(), target_a, atol=1e-2)
    assert torch.allclose(b.detach(), target_b, atol=1e-2)


def test_two_loop_fused_matches_serial_cpu():
    """The Gram-matrix two-loop (_two_loop_fused) must reproduce the
    serial reference recursion — CPU version of the GPU test, exercising
    the host-side algebra with the fallback multi_dot/lincomb."""
    import torch
    from fedkit.optim.lbfgsnew import LBFGSNew
    torch.manual_seed(5)
    n = 4099
    p = torch.nn.Parameter(torch.zeros(n))
    opt = LBFGSNew([p], history_size=7)
    g = torch.randn(n)
    hist_s = [torch.randn(n) for _ in range(6)]
    hist_y = [2.0 * s + 0.05 * torch.randn(n) for s in hist_s]
    SY = [[float(s.dot(y)) for y in hist_y] for s in hist_s]
    YY = [[float(a.dot(b)) for b in hist_y] for a in hist_y]
    H = 0.41
    d_ref = opt._two_loop(g, list(hist_y), list(hist_s), H,
                          [None] * 7, [None] * 7)
    d_fus = opt._two_loop_fused(g, hist_y, hist_s, H, SY, YY)
    assert torch.allclose(d_fus, d_ref, rtol=1e-5, atol=1e-5)
