"""Strategy-math property tests (SURVEY.md §4 implied test plan):
FedAvg of identical clients is a no-op; K=1 FedAvg == standalone; ADMM
z/y updates match the reference formulas; BB chain is deterministic."""

import math

import torch

from fedkit.algos import BBConfig, ConsensusADMM, FedAvg, FedProx
from fedkit.parallel.comm import LocalComm


def _mk(K=4, N=16, seed=0):
    torch.manual_seed(seed)
    comm = LocalComm(K, device=torch.device("cpu"))
    x = {ck: torch.randn(N) for ck in range(K)}
    return comm, x, N


def test_fedavg_mean_and_residual():
    comm, x, N = _mk()
    s = FedAvg()
    st = s.init_block(comm, N, "cpu")
    info = s.aggregate(comm, st, x, 0)
    mean = sum(x.values()) / comm.K
    assert torch.allclose(st["z"], mean, atol=1e-6)
    assert abs(info["dual"] - float(torch.norm(mean)) / N) < 1e-6


def test_fedavg_identical_clients_is_noop():
    comm, _, N = _mk()
    v = torch.randn(N)
    x = {ck: v.clone() for ck in range(comm.K)}
    s = FedAvg()
    st = s.init_block(comm, N, "cpu")
    s.aggregate(comm, st, x, 0)
    assert torch.allclose(st["z"], v, atol=1e-6)


def test_fedavg_k1_equals_standalone():
    comm = LocalComm(1, device=torch.device("cpu"))
    v = torch.randn(8)
    s = FedAvg()
    st = s.init_block(comm, 8, "cpu")
    s.aggregate(comm, st, {0: v.clone()}, 0)
    assert torch.allclose(st["z"], v)   # write-back restores the same params


def test_fedprox_penalty_and_residuals():
    comm, x, N = _mk()
    s = FedProx(rho0=2.0)
    st = s.init_block(comm, N, "cpu")
    xv = torch.randn(N, requires_grad=True)
    pen = s.penalty(st, 0, xv)
    assert torch.allclose(pen, 0.5 * 2.0 * (xv - st["z"]).norm() ** 2)
    info = s.aggregate(comm, st, x, 0)
    mean = sum(x.values()) / comm.K
    primal = sum(float(torch.norm(2.0 * (x[ck] - mean))) for ck in x) / N
    assert abs(info["primal"] - primal) < 1e-6


def test_admm_update_math():
    comm, x, N = _mk()
    rho = 0.3
    s = ConsensusADMM(rho0=rho)
    st = s.init_block(comm, N, "cpu")
    # seed duals
    for ck in range(comm.K):
        st["y"][ck] = torch.randn(N)
    y_before = {ck: st["y"][ck].clone() for ck in st["y"]}
    info = s.aggregate(comm, st, x, 0)
    znew = sum(y_before[ck] + rho * x[ck] for ck in x) / (comm.K * rho)
    assert torch.allclose(st["z"], znew, atol=1e-5)
    for ck in x:
        assert torch.allclose(st["y"][ck],
                              y_before[ck] + rho * (x[ck] - znew), atol=1e-5)
    assert info["rho"] == rho


def test_admm_penalty_formula():
    comm, _, N = _mk()
    s = ConsensusADMM(rho0=0.5)
    st = s.init_block(comm, N, "cpu")
    st["y"][1] = torch.randn(N)
    st["z"] = torch.randn(N)
    xv = torch.randn(N, requires_grad=True)
    pen = s.penalty(st, 1, xv)
    expect = st["y"][1].dot(xv - st["z"]) + 0.25 * (xv - st["z"]).norm() ** 2
    assert torch.allclose(pen, expect, atol=1e-5)


def test_bb_chain_deterministic_and_bounded():
    """BB rho chain: replayable from gathered scalars, rho stays < rhomax."""
    torch.manual_seed(7)
    comm = LocalComm(3, device=torch.device("cpu"))
    N = 32
    bb = BBConfig(enabled=True, period_T=1, alphacorrmin=0.0,
                  epsilon=1e-9, rhomax=0.1)
    s = ConsensusADMM(rho0=0.05, bb=bb)
    x0 = {ck: torch.randn(N) for ck in range(3)}
    st = s.init_block(comm, N, "cpu", x0)
    # round 0 just stores x0
    x = {ck: torch.randn(N) for ck in range(3)}
    s.aggregate(comm, st, x, 0)
    rho_after_0 = st["rho"]
    # round 1 runs the chain
    x1 = {ck: x[ck] + 0.1 * torch.randn(N) for ck in range(3)}
    s.aggregate(comm, st, x1, 1)
    assert st["rho"] < bb.rhomax + 1e-12
    assert not math.isnan(st["rho"])
    assert rho_after_0 == 0.05
