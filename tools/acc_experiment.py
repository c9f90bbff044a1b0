#!/usr/bin/env python3
"""Accuracy-ordering experiment — the reference's comparison.png protocol.

The reference's only published result is an accuracy figure (README.md:28-31,
K=10, `Net` CNN, CIFAR10): standalone K=1 (upper bound) >= FedAvg >=
consensus >= standalone K=10 (lower bound).  There is no network access for
the real CIFAR10, so this runs the SAME protocol on the deterministic
class-structured synthetic set (fedkit.data.cifar._synthetic_cifar) and
records the ordering.  Output: one JSON line per config on stdout; the
curated result is committed as profiles/acc_synthetic.md.

    python tools/acc_experiment.py [--nloop 4] [--nadmm 3] [--quick]
"""

import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fedkit.parallel import FedConfig, FederatedJob
from fedkit.parallel.runtime import run_standalone


def common(K, args, **kw):
    return FedConfig(
        K=K, model=args.model, default_batch=128,
        Nloop=args.nloop, Nepoch=1, Nadmm=args.nadmm,
        check_results=bool(getattr(args, "trajectory", False)),
        dtype=getattr(args, "dtype", "fp32"),
        save_model=False, load_model=False,
        init_model=True, biased_input=True, be_verbose=False,
        use_cuda=True, max_steps_per_epoch=args.max_steps, **kw)


def final_accs(job):
    return [100.0 * float(a) for a in job.verification_error_check()]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nloop", type=int, default=4)
    ap.add_argument("--nadmm", type=int, default=3)
    ap.add_argument("--max-steps", type=int, default=0, dest="max_steps")
    ap.add_argument("--quick", action="store_true",
                    help="tiny run for smoke testing")
    ap.add_argument("--only", type=str, default="",
                    help="comma list of configs to run")
    ap.add_argument("--rho", type=float, default=0.1)
    ap.add_argument("--bb", type=int, default=0)
    ap.add_argument("--warmup", type=int, default=0,
                    help="penalty warm-up rounds per block (FedProx/ADMM)")
    ap.add_argument("--jsonl-dir", type=str, default="",
                    help="write per-round residual JSONLs here")
    ap.add_argument("--trajectory", action="store_true",
                    help="evaluate every round (acc curves in the JSONLs)")
    ap.add_argument("--model", type=str, default="Net")
    ap.add_argument("--K", type=int, default=10)
    ap.add_argument("--dtype", type=str, default="fp32")
    args = ap.parse_args()
    if args.quick:
        args.nloop, args.nadmm, args.max_steps = 1, 1, 3

    results = {}
    t0 = time.time()
    only = [s for s in args.only.split(",") if s]

    def want(name):
        return not only or name in only

    if want("standalone_K1"):
        # standalone K=1, full data, Nepoch scaled to match total passes
        cfg = common(1, args)
        cfg.strategy = "none"
        cfg.Nepoch = args.nloop
        job = run_standalone(cfg)
        results["standalone_K1"] = final_accs(job)

    if want("standalone_K10"):
        cfg = common(args.K, args, strategy="none")
        if args.jsonl_dir:
            cfg.jsonl_path = os.path.join(args.jsonl_dir, "standalone10.jsonl")
        job = FederatedJob(cfg)
        job.run()
        results["standalone_K10"] = final_accs(job)

    if want("fedavg_K10"):
        cfg = common(args.K, args, strategy="fedavg")
        if args.jsonl_dir:
            cfg.jsonl_path = os.path.join(args.jsonl_dir, "fedavg.jsonl")
        job = FederatedJob(cfg)
        job.run()
        results["fedavg_K10"] = final_accs(job)

    if want("fedprox_K10"):
        cfg = common(args.K, args, strategy="fedprox", admm_rho0=args.rho,
                     penalty_warmup_rounds=args.warmup)
        if args.jsonl_dir:
            cfg.jsonl_path = os.path.join(args.jsonl_dir, "fedprox.jsonl")
        job = FederatedJob(cfg)
        job.run()
        results["fedprox_K10"] = final_accs(job)

    if want("admm_K10"):
        # consensus ADMM K=10 (rho0, optionally BB-adaptive)
        cfg = common(args.K, args, strategy="admm", admm_rho0=args.rho,
                     bb_update=bool(args.bb),
                     penalty_warmup_rounds=args.warmup)
        if args.jsonl_dir:
            cfg.jsonl_path = os.path.join(args.jsonl_dir, "admm.jsonl")
        job = FederatedJob(cfg)
        job.run()
        results["admm_K10"] = final_accs(job)

    out = {name: {"mean_acc": round(statistics.mean(a), 2),
                  "per_client": [round(v, 1) for v in a]}
           for name, a in results.items()}
    out["wall_s"] = round(time.time() - t0, 1)
    out["protocol"] = (f"{args.model}, K={args.K}, dtype={args.dtype}, "
                       f"synthetic class-structured CIFAR-shaped data, "
                       f"Nloop={args.nloop} Nadmm={args.nadmm} "
                       f"Nepoch=1, Adam lr=1e-3, batch 128")
    print(json.dumps(out))
    keys = ["standalone_K1", "fedavg_K10", "fedprox_K10", "admm_K10",
            "standalone_K10"]
    print("ordering", [(k, out[k]["mean_acc"]) for k in keys if k in out],
          file=sys.stderr)


if __name__ == "__main__":
    main()
