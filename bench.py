#!/usr/bin/env python3
"""fedkit flagship benchmark: ResNet18 FedAvg local-step throughput.

Measures the BASELINE.json headline ("CIFAR10 test acc + images/sec/client,
ResNet18 FedAvg K=8") throughput component on MI355X: each rank is one
federated client on one GPU running local training steps (fwd + CE loss +
bwd + Adam step, bf16 autocast, channels_last, hand-written CDNA4 kernels)
on synthetic CIFAR-shaped data, with the FedAvg parameter-subset all-reduce
(one layer-block, cycling, z written back) every AGG_EVERY steps inside the
timed region — a HIGHER communication cadence than the reference's
once-per-epoch (~49 steps at K=8), so comm cost is over- not under-counted.

    python bench.py --gpus N --steps K --warmup W
    # N>1 is launched by the driver via torch.distributed.run (one rank/GPU)

Prints ONE JSON line from rank 0: value = whole-job images/sec over all N
GPUs (weak scaling: per-GPU batch fixed at 128).
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

AGG_EVERY = 12
BATCH = 128


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch", type=int, default=BATCH)
    ap.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--native", type=int, default=1,
                    help="0 = eager torch ops for A/B comparison")
    ap.add_argument("--graph", type=int, default=0,
                    help="1 = hipGraph-capture the local step.  Measured "
                         "BOTH ways across boxes (r2: -6% on one box, +1.5% "
                         "on another with the run order reversed) — the "
                         "apparent gain tracks thermal run order, not the "
                         "graph; default off, kept for A/B")
    args = ap.parse_args()

    if not args.native:
        os.environ["FEDKIT_NATIVE"] = "0"

    import torch.distributed as dist
    from fedkit.models import ResNet18
    from fedkit.ops import flat as flat_ops
    from fedkit.ops import losses as L

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    # any torchrun launch (even nproc=1) goes through the full distributed
    # path — rendezvous, RCCL init, barriers, the FedAvg all-reduce — so a
    # 1-GPU lease exercises exactly the code the 8-GPU run will use
    distributed = world > 1 or "TORCHELASTIC_RUN_ID" in os.environ
    use_cuda = torch.cuda.is_available()
    if distributed:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")
        if use_cuda:
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    device = torch.device("cuda", torch.cuda.current_device()) if use_cuda \
        else torch.device("cpu")
    bf16 = args.dtype == "bf16" and use_cuda

    if use_cuda:
        import fedkit.ops
        if args.native and not fedkit.ops.has_ext():
            raise RuntimeError("fedkit._C not built — run __graft_entry__.build()")

    torch.manual_seed(1234 + rank)
    net = ResNet18().to(device)
    if use_cuda:
        net = net.to(memory_format=torch.channels_last)
    # fedkit FusedAdam: the whole update in ONE HIP kernel (torch's
    # foreach Adam spends ~110 us/step over ~10 multi-tensor sweeps;
    # torch's fused=True traded them for 1 kernel PLUS one step-counter
    # add per tensor — 62 launches, net loss).  Falls back to stock under
    # graph capture (the host-side step counter is not capture-safe).
    if use_cuda and args.native and not args.graph \
            and os.environ.get("FEDKIT_FUSED_ADAM", "1") != "0":
        from fedkit.optim import FusedAdam
        opt = FusedAdam(net.parameters(), lr=1e-3)
    else:
        opt = torch.optim.Adam(net.parameters(), lr=1e-3,
                               capturable=bool(args.graph and use_cuda))
    blocks = net.train_order_block_ids()
    params = list(net.parameters())

    # synthetic CIFAR-shaped pool resident on device (no network; BASELINE
    # config: random-init weights, synthetic data of the CIFAR10 shape)
    pool = 16
    xs = [torch.randn(args.batch, 3, 32, 32, device=device) for _ in range(pool)]
    if use_cuda:
        xs = [x.contiguous(memory_format=torch.channels_last) for x in xs]
    ys = torch.randint(0, 10, (pool, args.batch), device=device)

    def autocast():
        if bf16:
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
        import contextlib
        return contextlib.nullcontext()

    def fedavg_round(block_idx):
        """FedAvg one layer-block: all-reduce + mean + write back."""
        lo, hi = blocks[block_idx % len(blocks)]
        bp = [p.data for p in params[lo:hi + 1]]
        vec = flat_ops.pack(bp)
        if distributed:
            dist.all_reduce(vec)
            vec /= world
        flat_ops.unpack(vec, bp)

    step_i = 0

    def eager_step(x, y, keep_grad_buffers=False):
        # set_to_none=True matters eagerly: with persistent grad buffers
        # every step pays one fill + one accumulate-add kernel PER PARAM
        # (~120 launches, ~350 us on ResNet18); graph capture needs the
        # buffers kept
        opt.zero_grad(set_to_none=not keep_grad_buffers)
        with autocast():
            loss = L.cross_entropy(net(x), y)
        loss.backward()
        opt.step()

    # ---- hipGraph capture of the local step (MI355X: the inner loop is
    # launch-bound; one graph replay replaces ~380 host enqueues).  The
    # FedAvg all-reduce stays OUTSIDE the graph (collectives every
    # AGG_EVERY steps, eager).  Falls back to eager if capture fails.
    graph = None
    if use_cuda and args.graph:
        try:
            static_x = xs[0].clone()
            static_y = ys[0].clone()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    eager_step(static_x, static_y, keep_grad_buffers=True)
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                eager_step(static_x, static_y, keep_grad_buffers=True)
        except Exception as e:  # pragma: no cover - capture support varies
            print(f"[bench] hipGraph capture unavailable ({e}); eager path",
                  file=sys.stderr)
            graph = None
        else:
            print("[bench] hipGraph capture active", file=sys.stderr)

    def one_step():
        nonlocal step_i
        x = xs[step_i % pool]
        y = ys[step_i % pool]
        if graph is not None:
            static_x.copy_(x)
            static_y.copy_(y)
            graph.replay()
        else:
            eager_step(x, y)
        step_i += 1
        if step_i % AGG_EVERY == 0:
            fedavg_round(step_i // AGG_EVERY - 1)

    def barrier_sync():
        if distributed:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    if use_cuda:
        # clock-ramp preamble (setup, untimed, before the contracted
        # warmup): the FIRST process on a fresh box measures ~0.27 ms/step
        # slower than any later one at identical code (gpurun_out/
        # bench_pf_*.json) — GPU clocks ramp over the first ~1 s of load.
        a = torch.randn(4096, 4096, device=device, dtype=torch.bfloat16)
        t_ramp = time.perf_counter()
        while time.perf_counter() - t_ramp < 2.5:
            a = a @ a * 1e-3
        torch.cuda.synchronize()
        del a

    for _ in range(args.warmup):
        one_step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if use_cuda else world
    total_images = args.steps * args.batch * world
    value = total_images / elapsed
    ms_per_step = elapsed * 1000.0 / args.steps

    if rank == 0:
        print(json.dumps({
            "metric": "images_per_sec",
            "value": round(value, 1),
            "unit": "images/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if use_cuda else "fp32",
            "data": "synthetic CIFAR10-shaped (random-init weights; no network)",
            "config": {
                "model": "ResNet18",
                "global_batch": args.batch * world,
                "seq_len": None,
                "parallelism": f"fedavg_dp{world}",
                "local_batch": args.batch,
                "agg_every_steps": AGG_EVERY,
                "agg_payload": "one layer-block (param subset), cycling",
                "native_kernels": bool(args.native and use_cuda),
            },
        }))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
