"""Flagship benchmark: IMPALA CartPole-v1 learner step throughput on MI355X
(BASELINE.json metric: learner env-steps/sec at 1/2/4/8 GPUs).

Runs the full training iteration — fused model forward, V-trace, loss,
backward, flat-bucket RCCL all-reduce (N>1), fused clip+RMSprop — on
synthetic rollouts of the BASELINE config shape (obs 4, 2 actions, B=128,
S=5 per GPU; random-init MlpLSTMSingle, fp32: the reference's compute
dtype). Weak scaling: per-GPU batch fixed as N grows.

Contract (driver):
  python bench.py --gpus N --steps K --warmup W
  (N>1 is launched via torch.distributed.run, one rank per GPU over RCCL;
  rank 0 prints ONE JSON line.)

Secondary configs (BASELINE.json configs[2..5]) via --algo:
  --algo PPO | V-MPO | SAC | PPO-Continuous | SAC-Continuous
(continuous algos switch to the MountainCarContinuous shape: obs 2, 1 act).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent))

from pdrl_amd.agents.learner_module import switch_module  # noqa: E402
from pdrl_amd.parallel import GradReducer, init_distributed  # noqa: E402
from pdrl_amd.utils import load_params  # noqa: E402

OBS_DIM, N_ACTIONS = 4, 2  # CartPole-v1


def env_shape(algo: str):
    """(obs_dim, n_actions, continuous) of the BASELINE env for the algo."""
    if algo.endswith("Continuous"):
        return 2, 1, True  # MountainCarContinuous-v0
    return OBS_DIM, N_ACTIONS, False  # CartPole-v1


def make_synthetic_batch(params, device, seed):
    """Random rollout batch with the exact field shapes the learner consumes
    (synthetic data — no network access for real envs at benchmark scale)."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    B, S, H = params.batch_size, params.seq_len, params.hidden_size
    obs_dim, n_act, continuous = env_shape(params.algo)
    if continuous:
        acts = torch.tanh(torch.randn(B, S, n_act, generator=g))
        logits = torch.randn(B, S, 2 * n_act, generator=g)
        logp = -torch.rand(B, S, 1, generator=g)
    else:
        logits = torch.randn(B, S, n_act, generator=g)
        acts = torch.randint(0, n_act, (B, S, 1), generator=g).float()
        logp = torch.log_softmax(logits, dim=-1).gather(-1, acts.long())
    batch = {
        "obs": torch.randn(B, S, obs_dim, generator=g),
        "act": acts,
        "rew": torch.rand(B, S, 1, generator=g),
        "logits": logits,
        "log_prob": logp,
        "is_fir": (torch.rand(B, S, 1, generator=g) < 0.2).float(),
        "hx": torch.randn(B, S, H, generator=g) * 0.1,
        "cx": torch.randn(B, S, H, generator=g) * 0.1,
    }
    return {k: v.to(device) for k, v in batch.items()}


def _self_exec_torchrun(n: int):
    """--gpus N launched as a plain `python bench.py`: re-exec under
    torch.distributed.run so N ranks actually do the work. Without this,
    one rank would run and the whole-job aggregate below would overstate
    throughput N×."""
    import socket
    import subprocess

    with socket.socket() as s:  # free rendezvous port
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        f"--nproc-per-node={n}", "--master-addr=127.0.0.1",
        f"--master-port={port}", __file__, *sys.argv[1:],
    ]
    raise SystemExit(subprocess.call(cmd))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=50)
    ap.add_argument("--batch-size", type=int, default=128)
    ap.add_argument("--seq-len", type=int, default=5)
    ap.add_argument("--hidden", type=int, default=None,
                    help="hidden size (default: params.hidden_size=64; "
                         "64 runs the fully fused loss DAGs, other widths "
                         "use the fused cores + general loss path)")
    ap.add_argument("--mode", default="resident", choices=["resident", "staged"],
                    help="resident: batch lives on-device (kernel-DAG ceiling); "
                         "staged: ring drain + pinned H2D + step + weight "
                         "publish per iteration (system throughput)")
    ap.add_argument("--algo", default="IMPALA",
                    choices=["IMPALA", "PPO", "V-MPO", "SAC",
                             "PPO-Continuous", "SAC-Continuous"])
    args = ap.parse_args()

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        _self_exec_torchrun(args.gpus)

    rank, world = init_distributed()
    n_gpus = world  # the ranks actually doing work — never trust --gpus
    use_cuda = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    params = load_params()
    params.algo = args.algo
    params.batch_size = args.batch_size
    params.seq_len = args.seq_len
    if args.hidden is not None:
        params.hidden_size = args.hidden
    obs_dim, n_act, continuous = env_shape(args.algo)
    params.obs_dim, params.n_actions = obs_dim, n_act

    torch.manual_seed(1234)  # identical init across ranks
    updater_cls, model_cls = switch_module(args.algo)
    model = model_cls(obs_dim, n_act, params.seq_len, params.hidden_size)
    reducer = GradReducer() if world > 1 else None
    updater = updater_cls(model, params, device, grad_reducer=reducer)
    from pdrl_amd.ops.graphed import maybe_graph

    updater = maybe_graph(updater, device)

    if args.mode == "staged":
        # system throughput: per iteration = pinned-host fill + async H2D
        # staging + training step + actor weight publish (D2H + encode + TCP
        # send on a bound PUB — the reference publishes after every update,
        # ppo/learning.py:108)
        from pdrl_amd.agents.learner import BatchStager, WeightPublisher
        from pdrl_amd.transport import pub_bind
        from pdrl_amd.utils import Protocol, encode

        host_np = {k: v.numpy()
                   for k, v in make_synthetic_batch(params, "cpu", 100 + rank).items()}
        stager = BatchStager(device)
        pub = pub_bind("127.0.0.1", 35000 + 37 * rank) if rank == 0 else None
        actor = getattr(model, "actor", model)
        # Synchronous publisher on purpose: the threaded AsyncWeightPublisher
        # LOST 13 µs/step here to GIL contention (its 0.7 MB pickle blocks
        # this Python-driven hot loop); it stays the learner-process
        # default, where the loop has real idle gaps to absorb the thread.
        wpub = WeightPublisher(actor, device) if pub is not None else None

        # Software-pipelined iteration: the pinned-host fill and the
        # encode+TCP send overlap the GPU step. Every update's weights are
        # still broadcast (same wire traffic as the reference's
        # publish-after-every-update); each payload is SENT during the
        # following iteration, once its async D2H has completed.
        state = {"pending": False}

        def step_fn():
            dev_batch = stager.stage(host_np)   # fill overlaps prev GPU tail
            updater.step(dev_batch)
            if wpub is not None:
                if state["pending"]:
                    header, payload = encode(Protocol.Model, wpub.finish(),
                                             compress=False)
                    pub.send(header, payload)   # overlaps THIS step on GPU
                wpub.begin()                    # D2H queued after the step
                state["pending"] = True
    else:
        batch = make_synthetic_batch(params, device, seed=100 + rank)

        def step_fn():
            updater.step(batch)

    import torch.distributed as dist

    def barrier():
        if world > 1:
            dist.barrier()

    def sync():
        if use_cuda:
            torch.cuda.synchronize(device)

    for _ in range(args.warmup):
        step_fn()
    sync()
    barrier()
    sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        step_fn()
    sync()
    barrier()
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks (the job finishes when the slowest rank does)
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    transitions = n_gpus * args.batch_size * args.seq_len * args.steps
    value = transitions / elapsed
    if rank == 0:
        result = {
            "metric": "learner_env_steps_per_sec",
            "value": value,
            "unit": "env-steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": f"{type(model).__name__}(obs{obs_dim},act{n_act},H{params.hidden_size})",
                "algo": f"{args.algo} "
                        + ("MountainCarContinuous-v0" if continuous else "CartPole-v1"),
                "global_batch": args.batch_size * n_gpus,
                "seq_len": args.seq_len,
                "parallelism": f"dp{n_gpus}",
                "mode": args.mode,
            },
        }
        print(json.dumps(result))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
