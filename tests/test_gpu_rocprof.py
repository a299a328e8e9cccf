"""rocprof capture assertion: the CDNA4 warm-up/probe
kernels must actually dispatch on the GPU, verified from a rocprofv3
kernel trace rather than trusted from timings."""

import shutil
import sqlite3
import subprocess
import sys
from pathlib import Path

import pytest

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.timeout(300)
def test_probe_kernels_in_rocprof_trace(tmp_path):
    if shutil.which("rocprofv3") is None:
        pytest.skip("rocprofv3 not on PATH")
    out_dir = tmp_path / "prof"
    code = (
        "import sys; sys.path.insert(0, %r); "
        "import torch; torch.cuda.init(); "
        "from covalent_ssh_plugin_amd.gpu import probe; probe.warmup(0, 30)"
    ) % str(REPO)
    proc = subprocess.run(
        ["rocprofv3", "--kernel-trace", "-d", str(out_dir), "--", sys.executable, "-c", code],
        cwd="/tmp",
        env={"TMPDIR": "/tmp", **__import__("os").environ},
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    dbs = list(out_dir.rglob("*_results.db"))
    assert dbs, f"no rocprof results db under {out_dir}"
    names = set()
    for db in dbs:
        con = sqlite3.connect(db)
        for (table,) in con.execute(
            "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_info_kernel_symbol%'"
        ):
            for (display,) in con.execute(f"SELECT display_name FROM {table}"):
                names.add(display or "")
    joined = " ".join(names)
    assert "csp_mfma_spin_kernel" in joined, sorted(names)
    assert "csp_hbm_sweep" in joined, sorted(names)
