"""Single-box training run: spawns the real process topology via main.py
roles (learner + manager + worker group, separate OS processes) on loopback,
monitors the reward curve, and reports time-to-target.

    python scripts/local_train.py --algo IMPALA --workers 12 --minutes 5

This is BASELINE.json configs[0]/[1] measured end-to-end: real CartPole-v1
physics, real transport, real learner (HIP fused step when a GPU is
present).
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import socket
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def free_port_pair(excl=()):
    for _ in range(64):
        s1 = socket.socket()
        s1.bind(("127.0.0.1", 0))
        p = s1.getsockname()[1]
        s2 = socket.socket()
        try:
            s2.bind(("127.0.0.1", p + 1))
        except OSError:
            continue
        finally:
            s1.close()
            s2.close()
        if p not in excl and p + 1 not in excl:
            return p
    raise RuntimeError("no free port pair")


def read_reward_curve(result_dir: Path):
    curve = []
    f = result_dir / "scalars.jsonl"
    if not f.exists():
        return curve
    for line in f.read_text().splitlines():
        try:
            rec = json.loads(line)
        except json.JSONDecodeError:
            continue
        if rec.get("tag") == "50-game-mean-stat-of-epi-rew":
            curve.append((rec["wall"], rec["value"]))
    return curve


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--algo", default="IMPALA")
    ap.add_argument("--env", default="CartPole-v1")
    ap.add_argument("--workers", type=int, default=12)
    ap.add_argument("--minutes", type=float, default=5.0)
    ap.add_argument("--target", type=float, default=475.0)
    ap.add_argument("--lr", type=float, default=None)
    ap.add_argument("--entropy", type=float, default=None)
    ap.add_argument("--clip", type=float, default=None, help="max_grad_norm")
    ap.add_argument("--batch-size", type=int, default=None)
    ap.add_argument("--set", action="append", default=[],
                    help="extra param override key=value (repeatable)")
    args = ap.parse_args()

    from pdrl_amd.utils import load_params

    # write a params file for the child processes
    p = load_params()
    base = json.loads((REPO / "pdrl_amd/utils/parameters.json").read_text())
    base["algo"] = args.algo
    base["env"] = args.env
    if args.lr is not None:
        base["lr"] = args.lr
    if args.entropy is not None:
        base["entropy_coef"] = args.entropy
    if args.clip is not None:
        base["max_grad_norm"] = args.clip
    if args.batch_size is not None:
        base["batch_size"] = args.batch_size
    for kv in args.set:
        k, v = kv.split("=", 1)
        try:
            base[k] = json.loads(v)
        except json.JSONDecodeError:
            base[k] = v
    tmp = tempfile.NamedTemporaryFile("w", suffix=".json", delete=False)
    json.dump(base, tmp)
    tmp.close()
    env = dict(os.environ, PDRL_PARAMS=tmp.name, PYTHONPATH=str(REPO))

    mgr_port = free_port_pair()
    lrn_port = free_port_pair(excl=(mgr_port,))

    results_before = set((REPO / "results").glob("*")) if (REPO / "results").exists() else set()
    procs = []

    def spawn(*cmd):
        proc = subprocess.Popen([sys.executable, str(REPO / "main.py"), *map(str, cmd)],
                                env=env, cwd=REPO,
                                stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
        procs.append(proc)
        return proc

    # SIGTERM must run the finally-block child cleanup (timeout(1) sends it)
    def _on_term(*_):
        raise SystemExit(143)

    signal.signal(signal.SIGTERM, _on_term)

    t0 = time.monotonic()
    spawn("learner_sub_process", "127.0.0.1", lrn_port)
    spawn("manager_sub_process", "127.0.0.1", "127.0.0.1", mgr_port, lrn_port)
    spawn("worker_sub_process", args.workers, "127.0.0.1", "127.0.0.1", mgr_port, lrn_port)

    deadline = t0 + args.minutes * 60
    result_dir = None
    best, reached = -1e9, None
    try:
        while time.monotonic() < deadline:
            time.sleep(5.0)
            if result_dir is None:
                dirs = set((REPO / "results").glob("*")) - results_before \
                    if (REPO / "results").exists() else set()
                if dirs:
                    result_dir = sorted(dirs)[-1]
            if result_dir is None:
                continue
            curve = read_reward_curve(result_dir)
            if curve:
                rew = curve[-1][1]
                best = max(best, rew)
                el = time.monotonic() - t0
                print(f"[{el:6.1f}s] points={len(curve)} reward={rew:7.2f} "
                      f"best={best:7.2f}", flush=True)
                if rew >= args.target and reached is None:
                    reached = el
                    print(f"TARGET {args.target} reached in {reached:.1f}s")
                    break
            if any(pr.poll() is not None for pr in procs):
                print("a role process died; aborting")
                break
    finally:
        for pr in procs:
            try:
                pr.send_signal(signal.SIGTERM)
            except OSError:
                pass
        for pr in procs:
            try:
                pr.wait(10)
            except subprocess.TimeoutExpired:
                pr.kill()
    out = {
        "algo": args.algo, "env": args.env, "workers": args.workers,
        "target": args.target, "time_to_target_s": reached,
        "best_50_game_mean": best if best > -1e9 else None,
        "elapsed_s": time.monotonic() - t0,
        "result_dir": str(result_dir) if result_dir else None,
    }
    print(json.dumps(out))
    return 0 if reached is not None else 1


if __name__ == "__main__":
    sys.exit(main())
