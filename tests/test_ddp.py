"""Data-parallel learner tests over gloo, world_size 2 (CPU stand-in for the
RCCL/xGMI path — same torch.distributed code, different backend)."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _dist_worker(rank, world_size, port, fn_name, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from pdrl_amd.parallel import GradReducer, init_distributed

    init_distributed(backend="gloo")
    try:
        if fn_name == "allreduce":
            g1 = torch.full((10,), float(rank + 1))
            g2 = torch.arange(5, dtype=torch.float32) * (rank + 1)
            red = GradReducer()
            red.all_reduce([g1, g2])
            out_q.put((rank, g1.numpy(), g2.numpy()))
        elif fn_name == "learner_sync":
            from pdrl_amd.agents.learner_module import ImpalaUpdater
            from pdrl_amd.networks import MlpLSTMSingle
            from pdrl_amd.utils import load_params
            from tests.conftest import make_batch

            torch.manual_seed(123)  # same init on both ranks
            p = load_params()
            p.batch_size, p.seq_len, p.obs_dim, p.n_actions = 4, 5, 4, 2
            model = MlpLSTMSingle(4, 2, p.seq_len, p.hidden_size)
            red = GradReducer()
            red.broadcast_params(list(model.parameters()))
            upd = ImpalaUpdater(model, p, "cpu", grad_reducer=red)
            # DIFFERENT data per rank; identical updates expected after avg
            batch = make_batch(p, seed=1000 + rank)
            for _ in range(2):
                upd.step(batch)
            flat = torch.cat([q.detach().reshape(-1) for q in model.parameters()])
            out_q.put((rank, flat.numpy()))
    finally:
        dist.destroy_process_group()


def _run_dist(fn_name, world_size=2):
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    port = int(np.random.default_rng().integers(20000, 40000))
    procs = [
        ctx.Process(target=_dist_worker, args=(r, world_size, port, fn_name, out_q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    import queue as _q

    deadline = 120
    for _ in range(world_size):
        try:
            rank, *vals = out_q.get(timeout=deadline)
        except _q.Empty:
            break
        results[rank] = vals
    for p in procs:
        p.join(10)
        if p.is_alive():
            p.terminate()
    return results


def test_flat_allreduce_averages():
    res = _run_dist("allreduce")
    assert set(res) == {0, 1}
    for rank in (0, 1):
        g1, g2 = res[rank]
        np.testing.assert_allclose(g1, np.full(10, 1.5), rtol=1e-6)  # avg(1,2)
        np.testing.assert_allclose(g2, np.arange(5) * 1.5, rtol=1e-6)


def test_learner_ranks_stay_in_sync():
    """Two IMPALA learner ranks on different data end bit-identical after
    flat-gradient averaging."""
    res = _run_dist("learner_sync")
    assert set(res) == {0, 1}
    w0, w1 = res[0][0], res[1][0]
    np.testing.assert_allclose(w0, w1, rtol=1e-6, atol=1e-7)
    assert np.isfinite(w0).all()
