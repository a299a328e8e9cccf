"""Multi-rank bench readiness on CPU.

The driver launches ``bench.py`` under torch.distributed.run with one
rank per GPU.  No GPU exists in CI, so this verifies the N>1 path the
way the driver will drive it — 8 ranks over gloo on localhost — checking
rendezvous, the barrier-bracketed timing, the MAX-over-ranks reduction
and the whole-job aggregate math in the emitted JSON line.
"""

import json
import socket
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(600)
def test_bench_8rank_gloo_aggregate():
    port = _free_port()
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "8",
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            "bench.py", "--gpus", "8", "--steps", "8", "--warmup", "2",
            "--no-secondary",
        ],
        cwd=str(REPO),
        capture_output=True,
        text=True,
        timeout=570,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    json_lines = [
        line for line in proc.stdout.splitlines() if line.startswith("{")
    ]
    assert len(json_lines) == 1, (
        f"exactly ONE JSON line must be printed (rank 0), got "
        f"{len(json_lines)}: {proc.stdout[-2000:]}"
    )
    line = json.loads(json_lines[-1])
    assert line["n_gpus"] == 8
    assert line["steps"] == 8
    assert line["metric"] == "electrons_per_sec"
    assert line["scaling"] == "weak"
    assert line["higher_is_better"] is True
    # whole-job aggregate: value = steps * world_size / max-rank elapsed
    # and ms_per_step = elapsed / steps -> value * ms_per_step == 8000
    assert line["value"] > 0 and line["ms_per_step"] > 0
    implied_world = line["value"] * line["ms_per_step"] / 1000 * line["steps"] / 8
    assert abs(implied_world - 8) < 0.2, (
        f"aggregate math inconsistent: value={line['value']} "
        f"ms_per_step={line['ms_per_step']} -> implied world {implied_world}"
    )
    assert line["config"]["parallelism"] == "slots8"
    assert line["p50_ms"] > 0 and line["p99_ms"] >= line["p50_ms"]
