"""Client-to-client communication: RCCL over xGMI, or in-process.

The reference "communicates" by averaging tensors of K models living in one
process (federated_multi.py:203-217).  Here:

  LocalComm  — K simulated clients in one process; aggregation is a direct
               sum.  This IS the reference's execution model and is the CPU
               CI backend (the fake backend makes bit-level
               cross-checking natural).

  DistComm   — one process per client, one client per MI355X GPU.
               torch.distributed with the "nccl" backend (RCCL on ROCm);
               aggregation of a block's flat fp32 vector is ONE all-reduce
               over xGMI.  Block sizes are 456..4.7M fp32 (ResNet18
               partition), i.e. <= 19 MB messages: latency-dominated, so the
               collective runs on a dedicated comm stream and the engine
               overlaps it with the diagnostic forward where legal.

Scalar statistics (residual norms, BB-ADMM inner products) travel as a
single small all-gather ([K, n] fp64) so every rank computes identical
deterministic decisions.
"""

import datetime
import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class Communicator:
    """Abstract communicator over K federated clients."""

    K: int
    my_clients: List[int]

    @property
    def is_primary(self) -> bool:
        raise NotImplementedError

    def device(self) -> torch.device:
        raise NotImplementedError

    def sum_across_clients_async(self, per_client: Dict[int, torch.Tensor]):
        """Launch the cross-client sum; returns wait() -> summed tensor.

        The MI355X overlap hook: DistComm runs the RCCL
        all-reduce on a dedicated HIP stream so independent work (e.g. the
        per-round test-set evaluation for non-writeback strategies) proceeds
        on the compute stream; wait() fences the compute stream on the
        collective, without blocking the host.  Default: synchronous.
        """
        out = self.sum_across_clients(per_client)
        return lambda: out

    def sum_across_clients(self, per_client: Dict[int, torch.Tensor]) -> torch.Tensor:
        """Return sum over ALL K clients of a per-client vector.

        LocalComm: per_client has all K entries.  DistComm: per_client has
        this rank's single entry; the sum is an RCCL all-reduce.
        """
        raise NotImplementedError

    def gather_scalar_rows(self, per_client: Dict[int, List[float]]) -> torch.Tensor:
        """All-gather one row of scalars per client -> [K, n] fp64 on CPU."""
        raise NotImplementedError

    def barrier(self) -> None:
        pass


class LocalComm(Communicator):
    """All K clients simulated in this process (reference semantics)."""

    def __init__(self, K: int, device: Optional[torch.device] = None):
        self.K = K
        self.my_clients = list(range(K))
        self._device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")

    @property
    def is_primary(self) -> bool:
        return True

    def device(self) -> torch.device:
        return self._device

    def sum_across_clients(self, per_client):
        assert len(per_client) == self.K
        total = torch.zeros_like(per_client[self.my_clients[0]])
        for ck in sorted(per_client):
            total += per_client[ck]
        return total

    def gather_scalar_rows(self, per_client):
        rows = [per_client[ck] for ck in sorted(per_client)]
        return torch.tensor(rows, dtype=torch.float64)


class DistComm(Communicator):
    """One process per client over torch.distributed (RCCL on ROCm)."""

    def __init__(self, K: Optional[int] = None, backend: Optional[str] = None,
                 timeout_s: int = 600):
        if not dist.is_initialized():
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29517")
            dist.init_process_group(
                backend=backend,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        self.rank = dist.get_rank()
        world = dist.get_world_size()
        self.K = K if K is not None else world
        if self.K != world:
            raise ValueError(f"K={self.K} must equal world size {world} "
                             "(one process per client)")
        self.my_clients = [self.rank]
        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", self.rank))
            if os.environ.get("FEDKIT_FORCE_DEV0") == "1":
                # co-locate every rank on device 0 (test rig: exercises the
                # RCCL collectives with world>1 inside a 1-GPU lease)
                local_rank = 0
            torch.cuda.set_device(local_rank)
            self._device = torch.device("cuda", local_rank)
            # dedicated stream for collectives so the engine can overlap
            # aggregation with independent compute
            self.comm_stream = torch.cuda.Stream()
        else:
            self._device = torch.device("cpu")
            self.comm_stream = None

    @property
    def is_primary(self) -> bool:
        return self.rank == 0

    def device(self) -> torch.device:
        return self._device

    def sum_across_clients(self, per_client):
        assert list(per_client.keys()) == self.my_clients
        vec = per_client[self.rank].contiguous()
        dist.all_reduce(vec, op=dist.ReduceOp.SUM)
        return vec

    def sum_across_clients_async(self, per_client):
        assert list(per_client.keys()) == self.my_clients
        vec = per_client[self.rank].contiguous()
        if self.comm_stream is not None:
            # comm stream waits for the producer (pack) on the compute
            # stream, runs the collective, and wait() makes the compute
            # stream wait on the result -- no host blocking on either side
            ev = torch.cuda.Event()
            ev.record()
            with torch.cuda.stream(self.comm_stream):
                self.comm_stream.wait_event(ev)
                work = dist.all_reduce(vec, op=dist.ReduceOp.SUM,
                                       async_op=True)
            vec.record_stream(self.comm_stream)

            def wait():
                work.wait()      # fences the CURRENT (compute) stream
                return vec
            return wait
        work = dist.all_reduce(vec, op=dist.ReduceOp.SUM, async_op=True)

        def wait():
            work.wait()
            return vec
        return wait

    def gather_scalar_rows(self, per_client):
        row = torch.tensor(per_client[self.rank], dtype=torch.float64)
        if dist.get_backend() == "nccl":
            row_d = row.to(self._device)
            out = [torch.empty_like(row_d) for _ in range(self.K)]
            dist.all_gather(out, row_d)
            return torch.stack([o.cpu() for o in out])
        out = [torch.empty_like(row) for _ in range(self.K)]
        dist.all_gather(out, row)
        return torch.stack(out)

    def barrier(self):
        dist.barrier()


def make_comm(K: int, distributed: Optional[bool] = None,
              device: Optional[torch.device] = None) -> Communicator:
    """Pick DistComm when launched under torchrun (WORLD_SIZE set), else LocalComm."""
    if distributed is None:
        distributed = int(os.environ.get("WORLD_SIZE", "1")) > 1
    if distributed:
        return DistComm(K)
    return LocalComm(K, device=device)
