"""Diagnose worker-fleet scaling: N worker processes → one sink, aggregate
steps/s, with either a shared-memory model (deployment path) or per-process
private models.

    python scripts/worker_scale_bench.py --workers 8 [--private] [--seconds 10]
"""
from __future__ import annotations

import argparse
import os
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def _worker(model, idx, port, seconds, counter):
    import torch

    torch.set_num_threads(1)
    sys.path.insert(0, str(REPO))
    from pdrl_amd.agents import Worker
    from pdrl_amd.utils import load_params
    import main as main_mod

    p = load_params()
    p.algo, p.env = "IMPALA", "CartPole-v1"
    main_mod.probe_env_spaces(p)
    if model is None:
        model = main_mod.build_model(p).cpu().eval()
    try:
        aff = os.sched_getaffinity(0)
        if idx == 0:
            print(f"worker affinity: {len(aff)} cpus", flush=True)
    except AttributeError:
        pass
    w = Worker(model, idx, "127.0.0.1", port, "127.0.0.1", 1, p, seed=idx)
    deadline = time.monotonic() + seconds
    n = 0
    while time.monotonic() < deadline:
        w.collect(max_episodes=1)
        # count steps via local episode lengths is awkward; use heartbeat-free
        n += 1
    counter.value = n


def main():
    import torch
    import torch.multiprocessing as mp

    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--private", action="store_true",
                    help="per-process model instead of shared-memory model")
    ap.add_argument("--seconds", type=float, default=10.0)
    args = ap.parse_args()

    from pdrl_amd.transport import Endpoint
    from pdrl_amd.utils import load_params
    import main as main_mod

    ctx = mp.get_context("spawn")
    sink = Endpoint(bind=("127.0.0.1", 0), recv_hwm=1 << 20)

    model = None
    if not args.private:
        p = load_params()
        p.algo, p.env = "IMPALA", "CartPole-v1"
        main_mod.probe_env_spaces(p)
        model = main_mod.build_model(p).cpu().eval()
        model.share_memory()

    counters = [ctx.Value("q", 0) for _ in range(args.workers)]
    procs = [
        ctx.Process(target=_worker, args=(model, i, sink.bound_port, args.seconds, counters[i]))
        for i in range(args.workers)
    ]
    t0 = time.monotonic()
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(args.seconds + 60)
    dt = time.monotonic() - t0

    # drain sink and count steps (messages carry step CHUNKS)
    from pdrl_amd.utils import decode

    n = 0
    while True:
        msg = sink.recv(timeout=0.5)
        if msg is None:
            break
        _, data = decode(*msg)
        n += len(data) if isinstance(data, list) else 1
    eps = sum(c.value for c in counters)
    mode = "private" if args.private else "shared"
    print(f"{args.workers} workers ({mode}): {n} steps, {eps} episodes in "
          f"{dt:.1f}s → {n/dt:.0f} steps/s aggregate, {n/dt/args.workers:.0f}/worker")


if __name__ == "__main__":
    main()
