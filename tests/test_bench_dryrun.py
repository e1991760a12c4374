"""Multi-rank dry-run of the EXACT bench.py code path.

Launches bench.py under torch.distributed.run with the gloo backend and
--dry-run: compute is stubbed, but the collective code the driver's
8-GPU SCALE bench will execute — verdict all_reduce(MIN), subtree-root
all_gather + rank-0 cap finishing, barriers, max-over-ranks timing,
rank-0 JSON emit — runs verbatim at world sizes 2, 4 and 8.
"""
import json
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.parametrize("world", [2, 4, 8])
def test_bench_dry_run_world(world):
    port = 29580 + world
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={world}",
        "--master-addr=127.0.0.1",
        f"--master-port={port}",
        str(REPO / "bench.py"),
        "--gpus",
        str(world),
        "--steps",
        "2",
        "--warmup",
        "1",
        "--backend",
        "gloo",
        "--dry-run",
    ]
    out = subprocess.run(
        cmd, capture_output=True, text=True, timeout=300, cwd=str(REPO)
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [
        ln for ln in out.stdout.splitlines() if ln.startswith('{"metric"')
    ]
    assert len(lines) == 1, out.stdout[-2000:]
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == world
    assert rec["config"]["parallelism"] == f"shard{world}"
    assert rec["config"]["dry_run"] is True
    assert rec["steps"] == 2
