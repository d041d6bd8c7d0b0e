"""Actor-fleet ingest ceiling: N worker PROCESSES (each M vectorized envs,
C++ act + C++ batched env physics) collecting flat-out into local sinks.

    python scripts/fleet_bench.py [N] [--envs M] [--seconds S] [--algo A]

Measured end of round 2 on the MI355X box: 16 workers x 4 envs =
1.35 M env-steps/s aggregate (84 K/worker); 32 workers saturate the
box's cores at ~1.21 M. (Reference actor plane: <= 600 steps/s
cluster-wide — its workers sleep 50 ms per step by design.)
"""
from __future__ import annotations

import argparse
import multiprocessing as mp
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def run_worker(idx, seconds, envs, algo, env_name, counter):
    import threading

    import torch

    torch.set_num_threads(1)
    import main as main_mod
    from pdrl_amd.agents import Worker
    from pdrl_amd.transport import Endpoint
    from pdrl_amd.utils import load_params

    p = load_params()
    p.algo, p.env = algo, env_name
    p.num_envs_per_worker = envs
    main_mod.probe_env_spaces(p)
    model = main_mod.build_model(p)
    sink = Endpoint(bind=("127.0.0.1", 0))
    w = Worker(model, idx, "127.0.0.1", sink.bound_port, "127.0.0.1", 1, p,
               seed=idx)
    stop = threading.Event()
    w.stop_event = stop
    threading.Thread(target=lambda: (time.sleep(seconds), stop.set()),
                     daemon=True).start()
    w.collect()
    counter.value = w._total_steps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("workers", type=int, nargs="?", default=16)
    ap.add_argument("--envs", type=int, default=4)
    ap.add_argument("--seconds", type=float, default=8.0)
    ap.add_argument("--algo", default="IMPALA")
    ap.add_argument("--env", default="CartPole-v1")
    args = ap.parse_args()

    mp.set_start_method("spawn", force=True)
    counters = [mp.Value("l", 0) for _ in range(args.workers)]
    procs = [
        mp.Process(target=run_worker,
                   args=(i, args.seconds, args.envs, args.algo, args.env,
                         counters[i]))
        for i in range(args.workers)
    ]
    for p_ in procs:
        p_.start()
    for p_ in procs:
        p_.join(timeout=max(90.0, args.seconds * 4))
    total = sum(c.value for c in counters)
    rate = total / args.seconds
    print(f"{args.workers} workers x M={args.envs}: {total} steps in "
          f"{args.seconds:.0f}s -> {rate:.0f} steps/s aggregate "
          f"({rate / args.workers:.0f}/worker)")


if __name__ == "__main__":
    main()
