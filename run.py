"""Cluster launcher: generates tmux + ssh + rsync commands that start every
role on every machine listed in machines.json, then executes them.

Capability parity with the reference's run.py (tmux session per machine:
28-30; ssh connect: 32-34; repo rsync: 36-42; conda activate + python
main.py <role>: 44-52; learner/manager/worker command assembly: 54-95;
os.system execution: 98). Same entrypoint: ``python run.py``.
Pass ``--dry-run`` to print the commands without executing (new; testable).
"""
from __future__ import annotations

import os
import sys
from pathlib import Path

from pdrl_amd.utils import Machines

REPO = Path(__file__).resolve().parent
SESSION = "pdrl"
CONDA_ENV = os.environ.get("PDRL_CONDA_ENV", "")


def start_tmux_session(name: str) -> str:
    return f"tmux new-session -d -s {name}"


def new_window(session: str, window: str) -> str:
    return f"tmux new-window -t {session} -n {window}"


def send_keys(session: str, window: str, cmd: str) -> str:
    escaped = cmd.replace('"', '\\"')
    return f'tmux send-keys -t {session}:{window} "{escaped}" Enter'


def ssh_connect(ip: str) -> str:
    return f"ssh {ip}"


def copy_repo(ip: str) -> str:
    return f"rsync -az --exclude results --exclude logs --exclude gpurun_out {REPO}/ {ip}:{REPO}/"


def activate() -> str:
    return f"conda activate {CONDA_ENV} && " if CONDA_ENV else ""


def run_python(role: str, *args) -> str:
    arg_s = " ".join(str(a) for a in args)
    return f"cd {REPO} && {activate()}python main.py {role} {arg_s}"


def build_commands(machines=Machines) -> list[str]:
    cmds = [start_tmux_session(SESSION)]
    learner = machines.learner

    # learner machine
    cmds.append(new_window(SESSION, "learner"))
    chain = f"{ssh_connect(learner.ip)}" if learner.ip not in ("127.0.0.1", "localhost") else ""
    run = run_python("learner_sub_process", learner.ip, learner.port)
    cmds.append(send_keys(SESSION, "learner", f"{chain + ' ' if chain else ''}{run}".strip()))

    # one manager + one worker group per worker machine
    for i, w in enumerate(machines.workers):
        remote = w.ip not in ("127.0.0.1", "localhost")
        if remote:
            cmds.append(copy_repo(w.ip))
        win_m = f"manager-{i}"
        cmds.append(new_window(SESSION, win_m))
        run_m = run_python("manager_sub_process", w.manager_ip, learner.ip, w.port, learner.port)
        prefix = f"{ssh_connect(w.ip)} " if remote else ""
        cmds.append(send_keys(SESSION, win_m, f"{prefix}{run_m}"))

        win_w = f"worker-{i}"
        cmds.append(new_window(SESSION, win_w))
        run_w = run_python(
            "worker_sub_process", w.num_p, w.manager_ip, learner.ip, w.port, learner.port
        )
        cmds.append(send_keys(SESSION, win_w, f"{prefix}{run_w}"))
    return cmds


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cmds = build_commands()
    if "--dry-run" in argv:
        print("\n".join(cmds))
        return 0
    for c in cmds:
        rc = os.system(c)
        if rc != 0:
            print(f"[run.py] command failed ({rc}): {c}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
