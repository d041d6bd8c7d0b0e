"""bench.py driver-contract tests (CPU): the JSON line schema and the
multi-rank launch behavior the round-end SCALE run depends on."""
from __future__ import annotations

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _run_bench(*args, timeout=240):
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), *args],
        capture_output=True, text=True, cwd=REPO, timeout=timeout,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, out.stdout[-2000:]
    return json.loads(lines[-1])


def test_bench_single_rank_schema():
    rec = _run_bench("--steps", "3", "--warmup", "1")
    assert rec["metric"] == "learner_env_steps_per_sec"
    assert rec["n_gpus"] == 1
    assert rec["scaling"] == "weak"
    assert rec["higher_is_better"] is True
    assert rec["data"] == "synthetic"
    assert rec["config"]["parallelism"] == "dp1"
    assert rec["config"]["global_batch"] == 128
    # whole-job aggregate consistency: value == transitions / elapsed
    expect = rec["n_gpus"] * 128 * 5 / (rec["ms_per_step"] / 1000.0)
    assert abs(expect - rec["value"]) / rec["value"] < 1e-6


def test_bench_multirank_self_exec_reports_true_world():
    """`python bench.py --gpus 2` (no torchrun) must SELF-EXEC into 2 real
    ranks — the round-1 hazard was one rank doing the work while the line
    claimed a 2x aggregate. n_gpus/global_batch must reflect the actual
    world size."""
    rec = _run_bench("--gpus", "2", "--steps", "3", "--warmup", "1")
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 256


def test_bench_staged_mode_cpu():
    """--mode staged (pinned fill + step + pipelined weight publish) runs
    on CPU and reports the same schema with mode labeled."""
    rec = _run_bench("--mode", "staged", "--steps", "3", "--warmup", "1")
    assert rec["config"]["mode"] == "staged"
    assert rec["metric"] == "learner_env_steps_per_sec"
    assert rec["value"] > 0
