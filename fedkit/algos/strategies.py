"""Federated coordination strategies: FedAvg, FedProx, consensus ADMM (+BB).

Update math parity with the reference drivers:
  NoConsensus   -> no_consensus_multi.py (no communication, lower bound)
  FedAvg        -> federated_multi.py:203-217  (z = (1/K) sum x_k, z written
                   back into every model)
  FedProx       -> fedprox_multi.py:183-232    (closure adds (rho/2)||x-z||^2;
                   z = mean, NOT written back; primal/dual residuals)
  ConsensusADMM -> consensus_multi.py:193-302  (3-step ADMM with per-client
                   dual y_k, z = (1/(K rho)) sum(y_k + rho x_k),
                   y_k += rho (x_k - z); optional adaptive rho via
                   Barzilai-Borwein spectral stepsize, consensus_multi.py:241-278)

Distributed form (MI355X): "sum over clients" is ONE RCCL all-reduce of the
block's flat fp32 vector over xGMI; every scalar statistic is an all-gather
of a few doubles per rank, and every decision (BB rho chain) is replicated
deterministically on all ranks.  For BB the per-client inner products are
decomposed so the rho-dependent statistics can be evaluated for ANY rho from
6 gathered scalars:
    yhat_1 = (y - yhat0) + rho (x - z) = a + rho b
    d11 = a.a + 2 rho a.b + rho^2 b.b
    d12 = a.dx + rho b.dx            (dx = x - x0)
    d22 = dx.dx
which lets each rank replay the reference's sequential per-client rho chain
(where client ck's statistics see the rho already updated by clients < ck)
without moving any vectors.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch

from typing import TYPE_CHECKING
if TYPE_CHECKING:  # avoid the algos <-> parallel import cycle at runtime
    from ..parallel.comm import Communicator
else:
    Communicator = "Communicator"


@dataclass
class BBConfig:
    """Adaptive-rho (Barzilai-Borwein) knobs (consensus_multi.py:41-47)."""
    enabled: bool = False
    period_T: int = 2
    alphacorrmin: float = 0.2
    epsilon: float = 1e-3
    rhomax: float = 0.1


class Strategy:
    name = "base"
    writeback_z = False     # does aggregate() write z back into the models?
    uses_penalty = False    # does the closure add a penalty term?

    def init_block(self, comm: Communicator, N: int, device,
                   x0: Optional[Dict[int, torch.Tensor]] = None,
                   block_idx: int = 0) -> dict:
        """Fresh per-block state; z starts at 0 (federated_multi.py:151-154)."""
        return {"z": torch.zeros(N, dtype=torch.float32, device=device), "N": N}

    def penalty(self, state: dict, ck: int, xvec: torch.Tensor):
        """Extra (differentiable) loss term for client ck's closure."""
        return None

    def aggregate(self, comm: Communicator, state: dict,
                  x: Dict[int, torch.Tensor], round_idx: int) -> dict:
        """Run the communication round; returns log info.  Updates state."""
        raise NotImplementedError

    # -- split form: start() launches the collective (comm stream), finish()
    #    consumes it.  The engine can slot independent work (test-set eval
    #    for strategies that do not write z back) between the two.
    def aggregate_start(self, comm: Communicator, state: dict,
                        x: Dict[int, torch.Tensor]):
        return None

    def aggregate_finish(self, comm: Communicator, state: dict,
                         x: Dict[int, torch.Tensor], pending,
                         round_idx: int) -> dict:
        return self.aggregate(comm, state, x, round_idx)


class NoConsensus(Strategy):
    """K independent models, no communication ever (no_consensus_multi.py)."""
    name = "none"

    def aggregate(self, comm, state, x, round_idx):
        return {}


class FedAvg(Strategy):
    """Parameter-subset federated averaging (federated_multi.py)."""
    name = "fedavg"
    writeback_z = True

    def aggregate_start(self, comm, state, x):
        return comm.sum_across_clients_async(
            {k: v.clone() for k, v in x.items()})

    def aggregate_finish(self, comm, state, x, pending, round_idx):
        z, N = state["z"], state["N"]
        znew = pending()
        znew /= comm.K
        # device-side residual, NO host sync here: the engine enqueues the
        # writeback unpack (and the eval forwards behind it) immediately;
        # the float() happens at print/log time after everything is queued
        # (VERDICT r1 #10 — removes the host bubble between collective and
        # writeback on the FedAvg path)
        dual_residual = torch.norm(z - znew) / N
        state["z"] = znew
        return {"dual": dual_residual}

    def aggregate(self, comm, state, x, round_idx):
        return self.aggregate_finish(comm, state, x,
                                     self.aggregate_start(comm, state, x),
                                     round_idx)


class FedProx(Strategy):
    """FedAvg + proximal term (fedprox_multi.py); z is NOT written back."""
    name = "fedprox"
    uses_penalty = True

    def __init__(self, rho0: float = 1.0, warmup_rounds: int = 0):
        self.rho0 = rho0
        # engine extension, default 0 = reference behavior (see
        # ConsensusADMM.warmup_rounds: z starts at 0 per block, so the
        # round-0 proximal term pulls toward an uninformed target)
        self.warmup_rounds = warmup_rounds
        # per-(block, client) proximal weight table, lazily filled at rho0.
        # The reference allocates rho = torch.ones(L,3)*admm_rho0
        # (fedprox_multi.py:142, "per layer, per slave") but only ever READS
        # rho[ci,0] and never writes any entry, so every value stays rho0;
        # keeping the full table here implements the evident intent
        # (per-block, per-client weights) while remaining numerically
        # identical to the reference.
        self.rho_table: Dict[tuple, float] = {}

    def _rho(self, block_idx: int, ck: int) -> float:
        return self.rho_table.setdefault((block_idx, ck), self.rho0)

    def init_block(self, comm, N, device, x0=None, block_idx=0):
        st = super().init_block(comm, N, device)
        for ck in comm.my_clients:
            self._rho(block_idx, ck)
        st["block_idx"] = block_idx
        st["rho"] = self._rho(block_idx, comm.my_clients[0])
        return st

    def penalty(self, state, ck, xvec):
        if state.get("round", 0) < self.warmup_rounds:
            return None
        xdelta = xvec - state["z"]
        rho = self._rho(state.get("block_idx", 0), ck)
        return 0.5 * rho * (torch.norm(xdelta, 2) ** 2)

    def aggregate_start(self, comm, state, x):
        return comm.sum_across_clients_async(
            {k: v.clone() for k, v in x.items()})

    def aggregate(self, comm, state, x, round_idx):
        return self.aggregate_finish(comm, state, x,
                                     self.aggregate_start(comm, state, x),
                                     round_idx)

    def aggregate_finish(self, comm, state, x, pending, round_idx):
        z, N, rho = state["z"], state["N"], state["rho"]
        znew = pending()
        znew /= comm.K
        dual_residual = torch.norm(z - znew).item() / N
        state["z"] = znew
        # primal residual: sum_k ||rho (x_k - znew)|| / N  (fedprox_multi.py:228-232)
        per_client = {ck: [torch.norm(rho * (xv - znew)).item()]
                      for ck, xv in x.items()}
        rows = comm.gather_scalar_rows(per_client)
        primal_residual = float(rows[:, 0].sum()) / N
        state["round"] = round_idx + 1
        return {"dual": dual_residual, "primal": primal_residual, "rho": rho}


class ConsensusADMM(Strategy):
    """3-step consensus ADMM with optional BB adaptive rho (consensus_multi.py)."""
    name = "admm"
    uses_penalty = True

    def __init__(self, rho0: float = 0.1, bb: Optional[BBConfig] = None,
                 warmup_rounds: int = 0):
        self.rho0 = rho0
        self.bb = bb or BBConfig()
        # Engine extension (default 0 = reference behavior): skip the
        # augmented-Lagrangian penalty for the first `warmup_rounds`
        # communication rounds of each block.  Every block restarts with
        # z = 0 (consensus_multi.py:151-154), so round 0's penalty pulls
        # the block toward an UNINFORMED zero target; deferring it until z
        # is a real consensus (round >= 1) removes that damage without
        # changing the z-/y-updates (round-0 z = mean(x) regardless since
        # y = 0).  Measured on the synthetic protocol: see
        # profiles/acc_synthetic.md.
        self.warmup_rounds = warmup_rounds

    def init_block(self, comm, N, device, x0=None, block_idx=0):
        st = super().init_block(comm, N, device)
        st["rho"] = self.rho0
        st["y"] = {ck: torch.zeros(N, dtype=torch.float32, device=device)
                   for ck in comm.my_clients}
        if self.bb.enabled:
            # yhat0 starts at the block's initial parameter vector
            # (consensus_multi.py:171-180)
            st["yhat0"] = {ck: (x0[ck].clone() if x0 is not None else
                                torch.zeros(N, dtype=torch.float32, device=device))
                           for ck in comm.my_clients}
            st["x0"] = {ck: torch.zeros(N, dtype=torch.float32, device=device)
                        for ck in comm.my_clients}
        return st

    def penalty(self, state, ck, xvec):
        """y^T (x-z) + (rho/2)||x-z||^2  (consensus_multi.py:209-218)."""
        if state.get("round", 0) < self.warmup_rounds:
            return None
        xdelta = xvec - state["z"]
        return torch.dot(state["y"][ck], xdelta) \
            + 0.5 * state["rho"] * (torch.norm(xdelta, 2) ** 2)

    def _bb_update(self, comm, state, x, round_idx):
        """Replicated deterministic BB rho chain (consensus_multi.py:241-278)."""
        bb = self.bb
        z = state["z"]
        if round_idx == 0:
            for ck in comm.my_clients:
                state["x0"][ck] = x[ck].clone()
            return
        if round_idx % bb.period_T != 0:
            return
        # per-client scalars: a = y - yhat0, b = x - z, dx = x - x0
        per_client = {}
        for ck in comm.my_clients:
            a = state["y"][ck] - state["yhat0"][ck]
            b = x[ck] - z
            dx = x[ck] - state["x0"][ck]
            per_client[ck] = [float(a.dot(a)), float(a.dot(b)), float(b.dot(b)),
                              float(a.dot(dx)), float(b.dot(dx)), float(dx.dot(dx))]
        rows = comm.gather_scalar_rows(per_client)   # [K, 6] on every rank
        rho = state["rho"]
        rho_used = [0.0] * comm.K
        for ck in range(comm.K):
            aa, ab, bb_, adx, bdx, dxdx = (float(v) for v in rows[ck])
            rho_used[ck] = rho
            d11 = aa + 2 * rho * ab + rho * rho * bb_
            d12 = adx + rho * bdx
            d22 = dxdx
            if comm.is_primary:
                print('admm %d deltas=(%e,%e,%e)' % (round_idx, d11, d12, d22))
            rhonew = rho
            if abs(d12) > bb.epsilon and d11 > bb.epsilon and d22 > bb.epsilon:
                alpha = d12 / math.sqrt(d11 * d22)
                alphaSD = d11 / d22
                alphaMG = d12 / d22
                alphahat = alphaMG if 2.0 * alphaMG > alphaSD \
                    else alphaSD - 0.5 * alphaMG
                if alpha >= bb.alphacorrmin and alphahat < bb.rhomax:
                    rhonew = alphahat
                if comm.is_primary:
                    print('admm %d alphas=(%e,%e,%e)' % (round_idx, alpha, alphaSD, alphaMG))
            rho = rhonew
        state["rho"] = rho
        # carry forward own client's yhat/x0 with the rho seen at its turn
        for ck in comm.my_clients:
            state["yhat0"][ck] = state["y"][ck] + rho_used[ck] * (x[ck] - z)
            state["x0"][ck] = x[ck].clone()

    def aggregate(self, comm, state, x, round_idx):
        N = state["N"]
        if self.bb.enabled:
            self._bb_update(comm, state, x, round_idx)
        rho = state["rho"]
        z = state["z"]
        # z-update: (1/(K rho)) sum_k (y_k + rho x_k)  (consensus_multi.py:281-285)
        contrib = {ck: state["y"][ck] + rho * x[ck] for ck in x}
        znew = comm.sum_across_clients(contrib)
        znew /= (comm.K * rho)
        dual_residual = torch.norm(z - znew).item() / N
        state["z"] = znew
        # dual variable + primal residual (consensus_multi.py:291-296)
        per_client = {}
        for ck in x:
            ydelta = rho * (x[ck] - znew)
            per_client[ck] = [torch.norm(ydelta).item()]
            state["y"][ck].add_(ydelta)
        rows = comm.gather_scalar_rows(per_client)
        primal_residual = float(rows[:, 0].sum()) / N
        state["round"] = round_idx + 1
        return {"dual": dual_residual, "primal": primal_residual, "rho": rho}


STRATEGIES = {
    "none": NoConsensus,
    "fedavg": FedAvg,
    "fedprox": FedProx,
    "admm": ConsensusADMM,
}
