"""Single storage-shard ceiling: N workers publish packed chunks STRAIGHT
to one LearnerStorage (no manager in between), with the shard's
decode+assemble+ring loop profiled in-process.

    python scripts/shard_bench.py [--workers N] [--seconds S] [--profile]

This isolates the shard's own throughput from manager relay effects —
the tool that localized the round-2 ingest ceiling to trajectory
stacking (see docs/ROUND2_NOTES.md: 13K → 48K steps/s after the
numpy-end-to-end assembler).
"""
from __future__ import annotations

import argparse
import asyncio
import multiprocessing as mp
import socket
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(idx, port, seconds):
    import threading

    import torch

    torch.set_num_threads(1)
    sys.path.insert(0, str(REPO))
    import main as main_mod
    from pdrl_amd.agents import Worker
    from pdrl_amd.utils import load_params

    p = load_params()
    p.algo, p.env = "IMPALA", "CartPole-v1"
    p.num_envs_per_worker = 2
    main_mod.probe_env_spaces(p)
    model = main_mod.build_model(p)
    w = Worker(model, idx, "127.0.0.1", port, "127.0.0.1", 1, p, seed=idx)
    stop = threading.Event()
    w.stop_event = stop
    threading.Thread(target=lambda: (time.sleep(seconds), stop.set()),
                     daemon=True).start()
    w.collect()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=6)
    ap.add_argument("--seconds", type=float, default=8.0)
    ap.add_argument("--profile", action="store_true",
                    help="cProfile the shard loop (top lines to stdout)")
    args = ap.parse_args()

    mp.set_start_method("spawn", force=True)
    import torch

    torch.set_num_threads(1)
    import main as main_mod
    from pdrl_amd.agents import LearnerStorage
    from pdrl_amd.buffers import SharedRolloutRing, rollout_fields
    from pdrl_amd.utils import load_params

    p = load_params()
    p.algo, p.env = "IMPALA", "CartPole-v1"
    main_mod.probe_env_spaces(p)
    lport = free_port()
    fields = rollout_fields(p.obs_dim, p.n_actions, p.hidden_size, False)
    ring = SharedRolloutRing(fields, p.seq_len, 4096, True)
    s = LearnerStorage(ring, "127.0.0.1", lport, p)
    run_s = args.seconds + 8
    procs = [mp.Process(target=_worker, args=(i, lport, run_s))
             for i in range(args.workers)]
    for q in procs:
        q.start()
    time.sleep(4)  # spin-up

    async def run():
        tasks = [asyncio.create_task(s.ingest_task()),
                 asyncio.create_task(s.store_task())]
        n0 = s.n_ingested
        t0 = time.perf_counter()
        await asyncio.sleep(args.seconds)
        dt = time.perf_counter() - t0
        print(f"shard ingest {(s.n_ingested - n0) / dt:.0f} steps/s "
              f"(stored {s.n_stored} trajectories)")
        for t in tasks:
            t.cancel()

    if args.profile:
        import cProfile
        import pstats

        pr = cProfile.Profile()
        pr.enable()
        asyncio.run(run())
        pr.disable()
        pstats.Stats(pr).sort_stats("tottime").print_stats(14)
    else:
        asyncio.run(run())
    for q in procs:
        q.join(timeout=15)


if __name__ == "__main__":
    main()
