"""Megastep phase/barrier cost probe (GPU box only).

Prices (a) the grid barrier alone at several block counts, and (b) the
megastep kernel truncated after each phase (PDRL_MEGA_PHASE), to locate
where the single-launch step spends its time vs the multi-launch DAG.
"""
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

os.environ["PDRL_USE_GRAPH"] = "0"  # stream-ordered: env knob read per step


def sync():
    torch.cuda.synchronize()


def main():
    from pdrl_amd.ops import ext

    e = ext()
    dev = torch.device("cuda")
    bar = torch.zeros(1024, dtype=torch.int32, device=dev)
    for nb in (81, 128, 192, 256):
        e.barrier_bench(bar, nb, 10)
        sync()
        t0 = time.perf_counter()
        e.barrier_bench(bar, nb, 2000)
        sync()
        dt = (time.perf_counter() - t0) / 2000
        print(f"barrier nblocks={nb}: {dt * 1e6:.2f} us")

    from bench import env_shape, make_synthetic_batch  # noqa: E402
    from pdrl_amd.agents.learner_module import switch_module
    from pdrl_amd.utils import load_params

    params = load_params()
    params.algo = "IMPALA"
    params.obs_dim, params.n_actions, _ = env_shape("IMPALA")
    torch.manual_seed(0)
    upd_cls, model_cls = switch_module("IMPALA")

    os.environ["PDRL_MEGASTEP"] = "1"
    for phase in (1, 2, 3, 99):
        os.environ["PDRL_MEGA_PHASE"] = str(phase)
        model = model_cls(4, 2, params.seq_len, params.hidden_size)
        upd = upd_cls(model, params, dev)
        batch = make_synthetic_batch(params, dev, seed=7)
        for _ in range(200):
            upd.step(batch)
        sync()
        t0 = time.perf_counter()
        for _ in range(2000):
            upd.step(batch)
        sync()
        dt = (time.perf_counter() - t0) / 2000
        print(f"megastep max_phase={phase}: {dt * 1e6:.2f} us/step")

    os.environ["PDRL_MEGA_PHASE"] = "99"
    os.environ["PDRL_MEGASTEP"] = "0"
    model = model_cls(4, 2, params.seq_len, params.hidden_size)
    upd = upd_cls(model, params, dev)
    batch = make_synthetic_batch(params, dev, seed=7)
    for _ in range(200):
        upd.step(batch)
    sync()
    t0 = time.perf_counter()
    for _ in range(2000):
        upd.step(batch)
    sync()
    print(f"multi-launch DAG (no graph): {(time.perf_counter() - t0) / 2000 * 1e6:.2f} us/step")




def vmpo_phase_probe():
    """Truncate the V-MPO mega loss kernel after each phase to locate its
    cost (A catstats, B GAE, C search, C' softmax, D reductions, E grads)."""
    import os

    import torch
    from bench import env_shape, make_synthetic_batch
    from pdrl_amd.agents.learner_module import switch_module
    from pdrl_amd.utils import load_params

    dev = torch.device("cuda")
    params = load_params()
    params.algo = "V-MPO"
    params.obs_dim, params.n_actions, _ = env_shape("V-MPO")
    upd_cls, model_cls = switch_module("V-MPO")
    for phase in (1, 2, 31, 32, 33, 34, 3, 4, 5, 99):
        os.environ["PDRL_VMPO_PHASE"] = str(phase)
        torch.manual_seed(0)
        model = model_cls(4, 2, params.seq_len, params.hidden_size)
        upd = upd_cls(model, params, dev)
        batch = make_synthetic_batch(params, dev, seed=7)
        for _ in range(200):
            upd.step(batch)
        sync()
        t0 = time.perf_counter()
        for _ in range(2000):
            upd.step(batch)
        sync()
        print(f"vmpo max_phase={phase}: "
              f"{(time.perf_counter() - t0) / 2000 * 1e6:.2f} us/step")
    os.environ["PDRL_VMPO_PHASE"] = "99"


if __name__ == "__main__":
    main()
    vmpo_phase_probe()
