"""FederatedJob — the engine behind all driver entry points.

Factors the control flow that the reference copy-pastes across its six
drivers (federated_multi.py:143-220 and clones) into one
engine:

    for nloop in range(Nloop):                    # passes over the network
      for ci in range(L):                         # layer-blocks
        unfreeze block ci on every client
        fresh per-client optimizers; z = 0
        for nadmm in range(Nadmm):                # communication rounds
          for epoch in range(Nepoch):
            for ck in my_clients:                 # rank boundary (DistComm)
              for batch in shard(ck):             # local steps
                opt.step(closure)                 # fwd+loss+penalty+bwd
          x_k = flat trainable block  -> strategy.aggregate (RCCL all-reduce)
          optional z write-back; per-client eval; reference-format prints

With DistComm each client IS one rank pinned to one MI355X; with LocalComm
all K clients run sequentially in-process (the reference's execution model,
used for CPU CI and bit-level cross-checks).

bf16 mode (the MI355X fast path): channels_last activations, local step under
autocast-bf16 so conv/GEMM run the CDNA4 MFMA kernels on bf16 data while
parameters, aggregation, optimizer state and BN statistics stay fp32
(SURVEY.md §7 hard part 4).
"""

import contextlib
import json
import math
import os
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, Optional

import torch
import torch.optim as optim

from ..algos import BBConfig, ConsensusADMM, FedAvg, FedProx, NoConsensus, Strategy
from ..data import make_client_datasets
from ..models import MODEL_FACTORIES
from ..optim import FusedAdam, LBFGSNew
from ..ops import losses as loss_ops
from ..utils import (flat_trainable, freeze_all_layers, get_trainable_values,
                     init_weights, number_of_blocks, put_trainable_values,
                     save_client_checkpoint, load_client_checkpoint,
                     trainable_params, unfreeze_one_block, unfreeze_one_layer)
from .comm import Communicator, make_comm


@dataclass
class FedConfig:
    """Knob set with the reference's names and defaults
    (federated_multi.py:9-48, consensus_multi.py:9-59)."""
    K: int = 10
    default_batch: int = 128
    Nloop: int = 12
    Nepoch: int = 1
    Nadmm: int = 3
    lambda1: float = 0.0001
    lambda2: float = 0.0001
    admm_rho0: float = 0.1
    bb_update: bool = False
    bb_period_T: int = 2
    bb_alphacorrmin: float = 0.2
    bb_epsilon: float = 1e-3
    bb_rhomax: float = 0.1
    load_model: bool = False
    init_model: bool = True
    save_model: bool = True
    check_results: bool = True
    biased_input: bool = True
    be_verbose: bool = False
    use_resnet: bool = False
    use_cuda: bool = True
    # engine extensions (not in the reference)
    model: Optional[str] = None        # overrides use_resnet model choice
    strategy: str = "fedavg"           # none | fedavg | fedprox | admm
    optimizer: str = "adam"            # adam | lbfgs
    lr: float = 0.001
    seed: int = 69
    init_seed: int = 0
    dtype: str = "fp32"                # fp32 | bf16 (GPU fast path)
    per_layer: bool = False            # VAE driver freezes per layer, not per block
    diagnostic_forward: bool = True    # reference's second forward per step
    exact_reference_shards: bool = False
    data_root: str = "./torchdata"
    ckpt_prefix: str = "./s"
    jsonl_path: Optional[str] = None   # structured per-round metrics
    max_steps_per_epoch: int = 0       # 0 = full shard (tests/bench shrink it)
    max_eval_batches: int = 0          # 0 = full test set
    penalty_warmup_rounds: int = 0     # FedProx/ADMM: skip the penalty for
                                       # the first N rounds of each block
                                       # (z starts at 0 per block; 0 =
                                       # reference behavior)
    round_checkpoint: bool = False     # save s{k}.model + RNG sidecar at each
                                       # (nloop, ci) boundary; load_model=True
                                       # then RESUMES at the recorded round
                                       # instead of replaying (SURVEY §5
                                       # failure-recovery minimum)
    l2_all_blocks: bool = False        # VAE-CL: L2 reg on every block
                                       # (federated_vae_cl.py:230)

    def model_name(self) -> str:
        if self.model:
            return self.model
        return "ResNet18" if self.use_resnet else "Net"

    def make_strategy(self) -> Strategy:
        if self.strategy == "none":
            return NoConsensus()
        if self.strategy == "fedavg":
            return FedAvg()
        if self.strategy == "fedprox":
            return FedProx(rho0=self.admm_rho0,
                           warmup_rounds=self.penalty_warmup_rounds)
        if self.strategy == "admm":
            bb = BBConfig(self.bb_update, self.bb_period_T,
                          self.bb_alphacorrmin, self.bb_epsilon, self.bb_rhomax)
            return ConsensusADMM(rho0=self.admm_rho0, bb=bb,
                                 warmup_rounds=self.penalty_warmup_rounds)
        raise ValueError(f"unknown strategy {self.strategy!r}")


class FederatedJob:
    def __init__(self, cfg: FedConfig, comm: Optional[Communicator] = None,
                 model_factory: Optional[Callable] = None,
                 loss_fn: Optional[Callable] = None,
                 optimizer_factory: Optional[Callable] = None,
                 block_hook: Optional[Callable] = None):
        """loss_fn(net, inputs, labels) -> scalar loss (default cross-entropy).
        optimizer_factory(job, net, ci) -> optimizer (overrides cfg.optimizer;
        federated_vae_cl.py mixes Adam and LBFGS by block, 200-205).
        block_hook(job, ci) runs after the block is unfrozen (e.g. the VAE-CL
        reparametrization gate, federated_vae_cl.py:185-189)."""
        self.cfg = cfg
        self.optimizer_factory = optimizer_factory
        self.block_hook = block_hook
        self.last_running_loss = {}
        self.last_opts = {}
        torch.manual_seed(cfg.seed)
        self.comm = comm or make_comm(
            cfg.K, device=None if cfg.use_cuda else torch.device("cpu"))
        self.device = self.comm.device()
        if not cfg.use_cuda:
            self.device = torch.device("cpu")
        self.strategy = cfg.make_strategy()
        self.model_factory = model_factory or MODEL_FACTORIES[cfg.model_name()]
        self.loss_fn = loss_fn or (lambda net, x, y: loss_ops.cross_entropy(net(x), y))
        self.bf16 = (cfg.dtype == "bf16" and self.device.type == "cuda")
        self.channels_last = self.bf16 or (
            self.device.type == "cuda" and cfg.model_name().startswith("ResNet"))
        self._jsonl = open(cfg.jsonl_path, "a") if (
            cfg.jsonl_path and self.comm.is_primary) else None
        self._build_clients()

    # ------------------------------------------------------------------ setup

    def _build_clients(self):
        cfg = self.cfg
        self.nets: Dict[int, torch.nn.Module] = {}
        for ck in self.comm.my_clients:
            net = self.model_factory().to(self.device)
            if self.channels_last:
                net = net.to(memory_format=torch.channels_last)
            if cfg.load_model:
                load_client_checkpoint(net, ck, self.device, cfg.ckpt_prefix)
            self.nets[ck] = net
        if cfg.init_model and not cfg.load_model:
            for ck in self.comm.my_clients:
                # identical init on every client via a shared seed
                # (federated_multi.py:124-128) — replaces a broadcast
                torch.manual_seed(cfg.init_seed)
                self.nets[ck].apply(init_weights)
        self.train_loaders, self.test_loaders = make_client_datasets(
            cfg.K, self.comm.my_clients, cfg.default_batch, self.device,
            cfg.biased_input, cfg.data_root, cfg.exact_reference_shards,
            dtype=torch.float32, channels_last=self.channels_last,
            shuffle_seed=cfg.seed)
        net0 = self.nets[self.comm.my_clients[0]]
        if cfg.per_layer:
            from ..utils import number_of_layers
            self.Li = [[2 * i, 2 * i + 1]
                       for i in range(number_of_layers(net0) // 2)]
        else:
            self.Li = net0.train_order_block_ids()
        self.L = len(self.Li)

    def _make_optimizer(self, net, ci=-1):
        if self.optimizer_factory is not None:
            return self.optimizer_factory(self, net, ci)
        params = filter(lambda p: p.requires_grad, net.parameters())
        if self.cfg.optimizer == "lbfgs":
            return LBFGSNew(params, history_size=10, max_iter=4,
                            line_search_fn=True, batch_mode=True)
        # FusedAdam == torch Adam semantics/state-dict; one HIP kernel per
        # step on GPU, stock update elsewhere
        return FusedAdam(params, lr=self.cfg.lr)

    def _autocast(self):
        if self.bf16:
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
        return contextlib.nullcontext()

    # ------------------------------------------------------------------- eval

    def verification_error_check(self):
        """Per-client top-1 over the full test set; prints the reference's
        integer-floored accuracy line (no_consensus_multi.py:96-109)."""
        accs = {}
        for ck in self.comm.my_clients:
            net = self.nets[ck]
            net.eval()
            correct = torch.zeros((), dtype=torch.long, device=self.device)
            total = 0
            with torch.no_grad(), self._autocast():
                for bi, (images, labels) in enumerate(self.test_loaders[ck]):
                    if self.cfg.max_eval_batches and bi >= self.cfg.max_eval_batches:
                        break
                    outputs = net(images)
                    pred = outputs.float().argmax(dim=1)
                    correct += (pred == labels).sum()
                    total += labels.shape[0]
            net.train()
            accs[ck] = [float(correct.item()), float(total)]
        rows = self.comm.gather_scalar_rows(accs)
        if self.comm.is_primary:
            for ck in range(self.comm.K):
                correct, total = int(rows[ck, 0]), int(rows[ck, 1])
                print('Accuracy of the network %d on the %d test images:%%%f'
                      % (ck, total, 100 * correct // total))
        return rows[:, 0] / rows[:, 1]

    # ------------------------------------------------------------------ train

    def _local_epoch(self, ck, opt, state, ci, nloop, epoch):
        """One pass of client ck over its shard for the current block."""
        cfg = self.cfg
        net = self.nets[ck]
        running_loss = 0.0
        lin_ids = net.linear_layer_ids() if hasattr(net, "linear_layer_ids") else []
        for i, (inputs, labels) in enumerate(self.train_loaders[ck]):
            if cfg.max_steps_per_epoch and i >= cfg.max_steps_per_epoch:
                break

            def closure():
                if torch.is_grad_enabled():
                    opt.zero_grad()
                with self._autocast():
                    loss = self.loss_fn(net, inputs, labels)
                loss = loss.float()
                reg_on = (ci in lin_ids) or cfg.l2_all_blocks
                if self.strategy.uses_penalty or reg_on:
                    # PHYSICAL element order so the FedProx/ADMM penalty's
                    # (x - z) pairs elements the way the packed z was built
                    # (channels_last weights reorder under reshape(-1))
                    vec = flat_trainable(net)
                    pen = self.strategy.penalty(self._state, ck, vec)
                    if pen is not None:
                        loss = loss + pen
                    if reg_on:
                        # reference quirk kept: block index compared against
                        # parameter ids (federated_multi.py:183)
                        if cfg.lambda1:
                            loss = loss + cfg.lambda1 * torch.norm(vec, 1)
                        loss = loss + cfg.lambda2 * (torch.norm(vec, 2) ** 2)
                if loss.requires_grad:
                    loss.backward()
                return loss

            opt.step(closure)

            if cfg.diagnostic_forward:
                with torch.no_grad(), self._autocast():
                    loss1 = float(self.loss_fn(net, inputs, labels))
            else:
                loss1 = float("nan")
            running_loss += loss1
            if cfg.be_verbose and self.comm.is_primary:
                print('model=%d block=[%d,%d] %d(%d) minibatch=%d epoch=%d loss %e'
                      % (ck, self.Li[ci][0], self.Li[ci][1], nloop,
                         self._state["N"], i, epoch, loss1))
        return running_loss

    # -------------------------------------------------- per-round checkpoints

    def _round_state_path(self, ck):
        return f"{self.cfg.ckpt_prefix}{ck}.round"

    def _save_round_state(self, nloop, ci, opts):
        """Checkpoint at a (nloop, ci) boundary: the reference-contract
        s{k}.model file (federated_multi.py:226-233 layout, unchanged) plus
        a sidecar with everything needed to continue EXACTLY where the
        uninterrupted run would be — loader shuffle generators and global
        RNG (VAE reparametrization noise)."""
        for ck in self.comm.my_clients:
            save_client_checkpoint(
                self.nets[ck], opts.get(ck), self.cfg.Nepoch - 1,
                self.last_running_loss.get(ck, 0.0), ck, self.cfg.ckpt_prefix)
            side = {
                "nloop": nloop, "ci": ci,
                "loader_gen": self.train_loaders[ck].generator.get_state()
                if self.train_loaders[ck].generator is not None else None,
                "torch_rng": torch.get_rng_state(),
                "cuda_rng": torch.cuda.get_rng_state(self.device)
                if self.device.type == "cuda" else None,
                "running_loss": self.last_running_loss.get(ck, 0.0),
            }
            torch.save(side, self._round_state_path(ck))

    def _load_round_state(self):
        """Restore sidecar state; returns the last COMPLETED (nloop, ci)
        or None.  Model weights were already loaded by _build_clients
        (load_model=True path)."""
        marks = []
        for ck in self.comm.my_clients:
            path = self._round_state_path(ck)
            if not os.path.exists(path):
                return None
            side = torch.load(path, map_location="cpu", weights_only=False)
            marks.append((side["nloop"], side["ci"]))
            if side.get("loader_gen") is not None and \
                    self.train_loaders[ck].generator is not None:
                self.train_loaders[ck].generator.set_state(side["loader_gen"])
            torch.set_rng_state(side["torch_rng"])
            if side.get("cuda_rng") is not None and self.device.type == "cuda":
                torch.cuda.set_rng_state(side["cuda_rng"], self.device)
            self.last_running_loss[ck] = side.get("running_loss", 0.0)
        assert len(set(marks)) == 1, "clients disagree on resume round"
        return marks[0]

    def run(self):
        cfg = self.cfg
        comm = self.comm
        self.last_running_loss = {ck: 0.0 for ck in comm.my_clients}
        self.last_opts = {}
        done_upto = None
        if cfg.load_model and cfg.round_checkpoint:
            done_upto = self._load_round_state()
            if done_upto is not None and comm.is_primary:
                print('Resuming after loop=%d block=%d' % done_upto)
        for nloop in range(cfg.Nloop):
            for ci in range(self.L):
                if done_upto is not None and (nloop, ci) <= done_upto:
                    continue
                for ck in comm.my_clients:
                    if cfg.per_layer:
                        unfreeze_one_layer(self.nets[ck], ci)
                    else:
                        unfreeze_one_block(self.nets[ck], ci)
                if self.block_hook is not None:
                    self.block_hook(self, ci)
                net0 = self.nets[comm.my_clients[0]]
                N = sum(p.numel() for p in trainable_params(net0))
                x0 = None
                if cfg.bb_update:
                    x0 = {ck: get_trainable_values(self.nets[ck], self.device)
                          for ck in comm.my_clients}
                self._state = self.strategy.init_block(comm, N, self.device,
                                                       x0, block_idx=ci)
                opts = {ck: self._make_optimizer(self.nets[ck], ci)
                        for ck in comm.my_clients}
                self.last_opts = opts

                for nadmm in range(cfg.Nadmm):
                    t0 = time.perf_counter()
                    for epoch in range(cfg.Nepoch):
                        for ck in comm.my_clients:
                            self.last_running_loss[ck] = self._local_epoch(
                                ck, opts[ck], self._state, ci, nloop, epoch)
                    t_local = time.perf_counter() - t0

                    t0 = time.perf_counter()
                    x = {ck: get_trainable_values(self.nets[ck], self.device)
                         for ck in comm.my_clients}
                    # launch the all-reduce on the comm stream; for
                    # strategies that do not write z back (FedProx/ADMM the
                    # models keep their own x_k) the full test-set eval runs
                    # on the compute stream WHILE the collective is in
                    # flight over xGMI (SURVEY.md §5 overlap design)
                    pending = self.strategy.aggregate_start(
                        comm, self._state, x)
                    accs = None
                    if cfg.check_results and not self.strategy.writeback_z:
                        accs = self.verification_error_check()
                    info = self.strategy.aggregate_finish(
                        comm, self._state, x, pending, nadmm)
                    if self.strategy.writeback_z:
                        for ck in comm.my_clients:
                            put_trainable_values(self.nets[ck], self._state["z"])
                    t_comm = time.perf_counter() - t0

                    self._print_round(nloop, ci, nadmm, info,
                                      epoch=cfg.Nepoch - 1)
                    if cfg.check_results and accs is None:
                        accs = self.verification_error_check()
                    self._log_round(nloop, ci, nadmm, N, info, t_local,
                                    t_comm, accs)
                if cfg.round_checkpoint:
                    self._save_round_state(nloop, ci, opts)
        if comm.is_primary:
            print('Finished Training')
        if cfg.save_model:
            self.save_checkpoints()
        if self._jsonl:
            self._jsonl.close()

    # ------------------------------------------------------------- reporting

    def _print_round(self, nloop, ci, nadmm, info, epoch):
        if not self.comm.is_primary or not info:
            return
        lo, hi = self.Li[ci]
        if self.strategy.name == "fedavg":
            print('dual (epoch=%d,loop=%d,block=[%d,%d],avg=%d)=%e'
                  % (epoch, nloop, lo, hi, nadmm, info["dual"]))
        elif self.strategy.name in ("fedprox", "admm"):
            print('block=[%d,%d](%d,%f) ADMM=%d/%d primal=%e dual=%e'
                  % (lo, hi, self._state["N"], info.get("rho", 0.0),
                     nadmm, nloop, info["primal"], info["dual"]))

    def _log_round(self, nloop, ci, nadmm, N, info, t_local, t_comm, accs):
        if self._jsonl is None:
            return
        rec = {"nloop": nloop, "block": self.Li[ci], "N": N, "nadmm": nadmm,
               "t_local_s": round(t_local, 4), "t_comm_s": round(t_comm, 4),
               "bytes": 4 * N,
               **{k: (float(v) if isinstance(v, (int, float))
                      or torch.is_tensor(v) else v)
                  for k, v in info.items()}}
        if accs is not None:
            rec["acc"] = [round(float(a), 4) for a in accs]
        self._jsonl.write(json.dumps(rec) + "\n")
        self._jsonl.flush()

    def save_checkpoints(self):
        for ck in self.comm.my_clients:
            save_client_checkpoint(
                self.nets[ck], self.last_opts.get(ck),
                self.cfg.Nepoch - 1, self.last_running_loss.get(ck, 0.0),
                ck, self.cfg.ckpt_prefix)


def run_standalone(cfg: FedConfig, comm=None, model_factory=None, loss_fn=None):
    """no_consensus_multi.py semantics: per-epoch full-model training, no
    communication, fresh Adam per epoch, eval each epoch."""
    cfg.strategy = "none"
    job = FederatedJob(cfg, comm=comm, model_factory=model_factory, loss_fn=loss_fn)
    for ck in job.comm.my_clients:
        from ..utils import unfreeze_all_layers
        unfreeze_all_layers(job.nets[ck])
    for epoch in range(cfg.Nepoch):
        opts = {ck: job._make_optimizer(job.nets[ck]) for ck in job.comm.my_clients}
        job.last_opts = opts
        job._state = {"N": sum(p.numel() for p in job.nets[job.comm.my_clients[0]].parameters())}
        if job.comm.is_primary:
            print('Epoch %d' % epoch)
        for ck in job.comm.my_clients:
            job.last_running_loss[ck] = job._local_epoch(ck, opts[ck], job._state, -1, 0, epoch)
        if cfg.check_results:
            job.verification_error_check()
    if job.comm.is_primary:
        print('Finished Training')
    if cfg.save_model:
        job.save_checkpoints()
    return job
