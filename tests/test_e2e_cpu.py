"""End-to-end CPU runs of every driver path (tiny configs).

This is CI config #1 from BASELINE.json ("SimpleCNN K=2 no_consensus on
CPU") plus one tiny run per strategy and model family."""

import json
import math
import os

import pytest
import torch

from fedkit.parallel import FedConfig, FederatedJob
from fedkit.parallel.runtime import run_standalone


def tiny_cfg(tmp_path, **kw):
    base = dict(K=2, default_batch=32, Nloop=1, Nepoch=1, Nadmm=2,
                use_cuda=False, check_results=True, max_eval_batches=2,
                max_steps_per_epoch=2, save_model=True, be_verbose=False,
                ckpt_prefix=str(tmp_path / "s"),
                jsonl_path=str(tmp_path / "metrics.jsonl"))
    base.update(kw)
    return FedConfig(**base)


def test_no_consensus_cpu(tmp_path, capsys):
    cfg = tiny_cfg(tmp_path, strategy="none", Nepoch=2)
    run_standalone(cfg)
    out = capsys.readouterr().out
    assert "Accuracy of the network 0 on the" in out
    assert "Finished Training" in out
    assert os.path.exists(str(tmp_path / "s0.model"))
    assert os.path.exists(str(tmp_path / "s1.model"))


def test_fedavg_cpu_makes_clients_identical(tmp_path, capsys):
    cfg = tiny_cfg(tmp_path, strategy="fedavg")
    job = FederatedJob(cfg)
    job.run()
    out = capsys.readouterr().out
    assert "dual (epoch=0,loop=0,block=[" in out
    # after the final round of every block, z was written back into every
    # client and frozen blocks never move -> the K models are identical
    sd0 = job.nets[0].state_dict()
    sd1 = job.nets[1].state_dict()
    for k in sd0:
        assert torch.allclose(sd0[k], sd1[k], atol=1e-6), k
    # structured metrics log has one record per (block, round)
    recs = [json.loads(l) for l in open(tmp_path / "metrics.jsonl")]
    assert len(recs) == 5 * cfg.Nadmm          # Net has 5 blocks
    assert all("dual" in r and "t_comm_s" in r for r in recs)


def test_fedprox_cpu(tmp_path, capsys):
    cfg = tiny_cfg(tmp_path, strategy="fedprox", admm_rho0=1.0, check_results=False)
    FederatedJob(cfg).run()
    out = capsys.readouterr().out
    assert "primal=" in out and "dual=" in out


def test_admm_cpu_with_bb(tmp_path, capsys):
    cfg = tiny_cfg(tmp_path, strategy="admm", admm_rho0=0.1, Nadmm=3,
                   bb_update=True, bb_period_T=2, check_results=False)
    FederatedJob(cfg).run()
    out = capsys.readouterr().out
    assert "ADMM=" in out
    assert "deltas=" in out      # BB chain printed


def test_lbfgs_optimizer_path(tmp_path):
    cfg = tiny_cfg(tmp_path, strategy="fedavg", optimizer="lbfgs",
                   check_results=False, max_steps_per_epoch=1, save_model=False)
    job = FederatedJob(cfg)
    job.run()   # completes without error and produces finite params
    for p in job.nets[0].parameters():
        assert torch.isfinite(p).all()


def test_checkpoint_roundtrip(tmp_path):
    cfg = tiny_cfg(tmp_path, strategy="fedavg", check_results=False)
    job = FederatedJob(cfg)
    job.run()
    cfg2 = tiny_cfg(tmp_path, strategy="fedavg", load_model=True,
                    init_model=False, save_model=False, check_results=False)
    job2 = FederatedJob(cfg2)
    sd_saved = job.nets[0].state_dict()
    sd_loaded = job2.nets[0].state_dict()
    for k in sd_saved:
        assert torch.equal(sd_saved[k], sd_loaded[k])
    # the checkpoint dict has the reference's exact keys
    ck = torch.load(str(tmp_path / "s0.model"), weights_only=False)
    assert set(ck.keys()) == {"model_state_dict", "epoch",
                              "optimizer_state_dict", "running_loss"}


@pytest.mark.slow
def test_resnet9_one_block_cpu(tmp_path):
    cfg = tiny_cfg(tmp_path, model="ResNet9", strategy="fedavg", Nadmm=1,
                   default_batch=8, check_results=False, save_model=False)
    cfg.Nloop = 1
    job = FederatedJob(cfg)
    job.run()
    for p in job.nets[0].parameters():
        assert torch.isfinite(p).all()


def test_admm_dual_residual_decreases(tmp_path):
    """ADMM health — the dual residual ||z - z_new||/N shrinks
    over communication rounds within a block (z converging)."""
    import json as _json
    from collections import defaultdict
    cfg = tiny_cfg(tmp_path, strategy="admm", admm_rho0=0.1, Nadmm=3,
                   check_results=False, max_steps_per_epoch=4)
    FederatedJob(cfg).run()
    rows = [_json.loads(line)
            for line in open(tmp_path / "metrics.jsonl")]
    by_block = defaultdict(list)
    for r in rows:
        by_block[tuple(r["block"])].append(r["dual"])
    assert by_block
    improved = sum(1 for v in by_block.values() if v[-1] <= v[0])
    assert improved >= 0.7 * len(by_block), dict(by_block)


def test_fedprox_primal_residual_logged(tmp_path):
    import json as _json
    cfg = tiny_cfg(tmp_path, strategy="fedprox", admm_rho0=0.1,
                   check_results=False)
    FederatedJob(cfg).run()
    rows = [_json.loads(line)
            for line in open(tmp_path / "metrics.jsonl")]
    assert all("primal" in r and math.isfinite(r["primal"]) for r in rows)


def test_round_checkpoint_kill_and_resume(tmp_path):
    """SURVEY §5 failure-recovery: checkpoint at every (nloop, ci) boundary;
    a killed run resumed with load_model=True reproduces the uninterrupted
    run's final weights, z and logged residuals exactly."""

    def cfg_for(sub, **kw):
        d = tmp_path / sub
        d.mkdir(exist_ok=True)
        return FedConfig(K=2, default_batch=32, Nloop=2, Nepoch=1, Nadmm=2,
                         use_cuda=False, check_results=False,
                         max_steps_per_epoch=2, save_model=False,
                         strategy="admm", admm_rho0=0.5, model="Net",
                         round_checkpoint=True,
                         ckpt_prefix=str(d / "s"),
                         jsonl_path=str(d / "metrics.jsonl"), **kw)

    # uninterrupted reference run
    job_a = FederatedJob(cfg_for("a"))
    job_a.run()
    sd_a = {ck: job_a.nets[ck].state_dict() for ck in (0, 1)}
    z_a = job_a._state["z"].clone()
    recs_a = [json.loads(l) for l in open(tmp_path / "a" / "metrics.jsonl")]

    # interrupted run: kill at the start of (nloop=1, ci=1)
    class Killed(Exception):
        pass

    def killer(job, ci):
        if job._kill_at == (job._cur_nloop, ci):
            raise Killed

    job_b = FederatedJob(cfg_for("b"), block_hook=killer)
    job_b._kill_at = (1, 1)
    # the hook infers nloop by watching ci wrap around
    nl = {"v": 0, "seen": -1}

    def hook(job, ci):
        if ci <= nl["seen"]:
            nl["v"] += 1
        nl["seen"] = ci
        job._cur_nloop = nl["v"]
        killer(job, ci)
    job_b.block_hook = hook
    with pytest.raises(Killed):
        job_b.run()

    # resume from the checkpoint and finish
    job_c = FederatedJob(cfg_for("b", load_model=True))
    job_c.run()
    sd_c = {ck: job_c.nets[ck].state_dict() for ck in (0, 1)}
    z_c = job_c._state["z"]

    for ck in (0, 1):
        for k in sd_a[ck]:
            assert torch.allclose(sd_a[ck][k].float(), sd_c[ck][k].float(),
                                  atol=1e-7), (ck, k)
    assert torch.allclose(z_a, z_c, atol=1e-7)
    # the resumed run's logged residuals match the uninterrupted tail
    recs_bc = [json.loads(l) for l in open(tmp_path / "b" / "metrics.jsonl")]
    tail_a = [r for r in recs_a if (r["nloop"], tuple(r["block"])) >= (1,)
              and r["nloop"] == 1]
    tail_c = [r for r in recs_bc if r["nloop"] == 1 and r.get("resumed", True)]
    done = [r for r in tail_c if tuple(r["block"]) == tuple(tail_a[-1]["block"])]
    assert done, "resumed run never reached the final block"
    assert abs(done[-1]["dual"] - tail_a[-1]["dual"]) < 1e-9
    assert abs(done[-1]["primal"] - tail_a[-1]["primal"]) < 1e-9


def test_acc_experiment_tool_quick(tmp_path, monkeypatch):
    """Guard the accuracy-protocol tool against bitrot (CPU quick mode)."""
    import subprocess
    import sys
    r = subprocess.run(
        [sys.executable, "tools/acc_experiment.py", "--quick",
         "--only", "fedavg_K10", "--model", "Net", "--K", "2"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-500:]
    rec = json.loads([l for l in r.stdout.splitlines()
                      if l.startswith("{")][-1])
    assert "fedavg_K10" in rec and rec["fedavg_K10"]["mean_acc"] > 0


def test_sync_debug_wrapper():
    """FEDKIT_SYNC_DEBUG wraps native calls with a post-call check."""
    from fedkit.ops import _SyncDebugExt

    class FakeMod:
        const = 7

        def op(self, a):
            return a + 1

    w = _SyncDebugExt(FakeMod())
    assert w.const == 7
    assert w.op(1) == 2   # torch.cuda.synchronize is a no-op without GPU


def test_round_checkpoint_resume_vae_rng(tmp_path):
    """Resume reproducibility for an RNG-consuming model (VAE
    reparametrization draws from the global torch RNG, which the sidecar
    saves/restores)."""
    from fedkit.ops.losses import vae_loss

    def loss_fn(net, images, _labels):
        out, mu, logvar = net(images)
        return vae_loss(out, images, mu, logvar)

    def cfg_for(sub, **kw):
        d = tmp_path / sub
        d.mkdir(exist_ok=True)
        return FedConfig(K=1, default_batch=16, Nloop=2, Nepoch=1, Nadmm=1,
                         use_cuda=False, check_results=False,
                         max_steps_per_epoch=1, save_model=False,
                         strategy="fedavg", model="AutoEncoderCNN",
                         per_layer=True, round_checkpoint=True,
                         ckpt_prefix=str(d / "s"), **kw)

    job_a = FederatedJob(cfg_for("a"), loss_fn=loss_fn)
    job_a.run()
    sd_a = {k: v.clone() for k, v in job_a.nets[0].state_dict().items()}

    class Killed(Exception):
        pass

    nl = {"v": 0, "seen": -1}

    def hook(job, ci):
        if ci <= nl["seen"]:
            nl["v"] += 1
        nl["seen"] = ci
        if (nl["v"], ci) == (1, 3):
            raise Killed

    job_b = FederatedJob(cfg_for("b"), loss_fn=loss_fn, block_hook=hook)
    with pytest.raises(Killed):
        job_b.run()
    job_c = FederatedJob(cfg_for("b", load_model=True), loss_fn=loss_fn)
    job_c.run()
    sd_c = job_c.nets[0].state_dict()
    for k in sd_a:
        assert torch.allclose(sd_a[k], sd_c[k], atol=1e-7), k


@pytest.mark.slow
def test_cpc_driver_cpu_end_to_end(tmp_path):
    """federated_cpc.py end to end on CPU: synthetic LOFAR visibilities,
    LBFGS closures, InfoNCE, per-sub-model FedAvg, reference print
    format (the driver the reference cannot even start — its
    unfreeze_one_block TypeError — runs here)."""
    import subprocess
    import sys
    r = subprocess.run(
        [sys.executable, "federated_cpc.py", "--K", "2", "--Niter", "1",
         "--Nloop", "1", "--Nadmm", "1", "--save_model", "0",
         "--load_model", "0", "--use_cuda", "0"],
        capture_output=True, text=True, timeout=540,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-800:]
    assert "dual (N=" in r.stdout     # reference-format FedAvg residual
