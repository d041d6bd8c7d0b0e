"""End-to-end ingest bench: N workers → manager → storage → shared ring,
no learner. Reports steps ingested and trajectories stored per second, plus
per-stage counters to localize bottlenecks.

    python scripts/ingest_bench.py --workers 8 --seconds 15
"""
from __future__ import annotations

import argparse
import socket
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def free_port(span: int = 0) -> int:
    """A port p with p..p+span all bindable (shards use p+2+k)."""
    for _ in range(128):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        socks = []
        try:
            for off in range(span + 1):
                t = socket.socket()
                t.bind(("127.0.0.1", p + off))
                socks.append(t)
        except OSError:
            continue
        finally:
            for t in socks:
                t.close()
        return p
    raise RuntimeError("no free port span")


def _worker(model, idx, mport, seconds):
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
    w = Worker(model, idx, "127.0.0.1", mport, "127.0.0.1", 1, p, seed=idx)
    deadline = time.monotonic() + seconds
    while time.monotonic() < deadline:
        w.collect(max_episodes=1)


def _manager(mport, lport, seconds, relayed, shards=1, drops=None):
    sys.path.insert(0, str(REPO))
    import torch

    torch.set_num_threads(1)
    from pdrl_amd.agents import Manager

    m = Manager("127.0.0.1", mport, "127.0.0.1", lport, storage_shards=shards)
    deadline = time.monotonic() + seconds
    n = 0
    while time.monotonic() < deadline:
        k = m.relay_batch(timeout=0.2)
        if k:
            n += k
            relayed.value = n
        if drops is not None:
            # [rx drops at manager sub, tx drops across shard pubs]
            drops[0] = m.sub.stats()["rx_dropped"]
            drops[1] = sum(p.stats()["tx_dropped"] for p in m.pubs)


def _storage(ring, lport, seconds, ingested, stored):
    sys.path.insert(0, str(REPO))
    import asyncio
    import os

    import torch

    torch.set_num_threads(1)
    from pdrl_amd.agents import LearnerStorage
    from pdrl_amd.utils import load_params
    import main as main_mod

    p = load_params()
    p.algo, p.env = "IMPALA", "CartPole-v1"
    main_mod.probe_env_spaces(p)
    s = LearnerStorage(ring, "127.0.0.1", lport, p)

    async def run():
        t = asyncio.get_event_loop().time
        deadline = t() + seconds
        tasks = [asyncio.create_task(s.ingest_task()),
                 asyncio.create_task(s.store_task())]
        while t() < deadline:
            await asyncio.sleep(0.5)
            ingested.value = s.n_ingested
            stored.value = s.n_stored
        for task in tasks:
            task.cancel()

    if os.environ.get("PDRL_PROF_STORAGE"):
        import cProfile
        import pstats

        pr = cProfile.Profile()
        pr.enable()
        asyncio.run(run())
        pr.disable()
        with open("/tmp/pdrl_storage_prof.txt", "w") as fh:
            pstats.Stats(pr, stream=fh).sort_stats("tottime").print_stats(16)
    else:
        asyncio.run(run())


def main():
    import torch.multiprocessing as mp

    from pdrl_amd.buffers import SharedRolloutRing, rollout_fields
    from pdrl_amd.utils import load_params
    import main as main_mod

    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--shards", type=int, default=1,
                    help="storage shard processes (manager routes by worker)")
    ap.add_argument("--seconds", type=float, default=15.0)
    ap.add_argument("--shared-model", action="store_true")
    args = ap.parse_args()

    p = load_params()
    p.algo, p.env = "IMPALA", "CartPole-v1"
    main_mod.probe_env_spaces(p)
    from pdrl_amd.agents import storage_shard_ports

    ctx = mp.get_context("spawn")
    mport, lport = free_port(), free_port(2 + args.shards)
    fields = rollout_fields(p.obs_dim, p.n_actions, p.hidden_size, False)
    # big off-policy ring so storage never blocks on a full ring
    ring = SharedRolloutRing(fields, p.seq_len, 65536, on_policy=False)

    model = None
    if args.shared_model:
        model = main_mod.build_model(p).cpu().eval()
        model.share_memory()

    relayed = ctx.Value("q", 0)
    drops = ctx.Array("q", 2)
    ingesteds = [ctx.Value("q", 0) for _ in range(args.shards)]
    storeds = [ctx.Value("q", 0) for _ in range(args.shards)]
    run_s = args.seconds + 10  # children outlive the measure window
    procs = [ctx.Process(target=_manager,
                         args=(mport, lport, run_s, relayed, args.shards, drops))]
    procs += [ctx.Process(target=_storage,
                          args=(ring, port, run_s, ingesteds[k], storeds[k]))
              for k, port in enumerate(storage_shard_ports(lport, args.shards))]
    procs += [ctx.Process(target=_worker, args=(model, i, mport, run_s))
              for i in range(args.workers)]
    for pr in procs:
        pr.start()
    time.sleep(5)  # spin-up
    r0 = relayed.value
    i0 = sum(c.value for c in ingesteds)
    s0 = sum(c.value for c in storeds)
    t0 = time.monotonic()
    time.sleep(args.seconds)
    dt = time.monotonic() - t0
    r1 = relayed.value
    i1 = sum(c.value for c in ingesteds)
    s1 = sum(c.value for c in storeds)
    print(f"workers={args.workers} shards={args.shards} "
          f"relayed={(r1-r0)/dt:8.0f}/s "
          f"ingested={(i1-i0)/dt:8.0f}/s stored={(s1-s0)/dt:8.0f} traj/s "
          f"(~{(s1-s0)*p.seq_len/dt:.0f} steps/s) "
          f"mgr_drops: rx={drops[0]} tx={drops[1]} (cumulative)")
    for pr in procs:
        pr.terminate()
    for pr in procs:
        pr.join(5)


if __name__ == "__main__":
    main()
