"""bench.py contract tests (CPU tier).

The driver runs ``python bench.py --gpus N ...`` (torchrun for N>1) and
parses ONE JSON line from rank 0.  These tests run the real script at a
tiny configuration -- single process and gloo world_size=2 -- so the
distributed code path the driver exercises on 8 GPUs is covered here
without a GPU.
"""

import json
import os
import socket
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
BENCH = str(REPO / "bench.py")

TINY = ["--rows", "2000", "--features", "64", "--cands-per-step", "4",
        "--epochs", "2", "--steps", "1", "--warmup", "0", "--cv", "3",
        "--cpu"]

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling",
                 "vs_baseline", "dtype", "data", "config"}


def _parse_json_line(stdout):
    lines = [ln for ln in stdout.splitlines() if ln.startswith("{")]
    assert lines, f"no JSON line in output:\n{stdout}"
    return json.loads(lines[-1])


@pytest.mark.timeout(300)
def test_bench_single_process_contract():
    out = subprocess.run(
        [sys.executable, BENCH] + TINY, capture_output=True, text=True,
        timeout=240, cwd=str(REPO))
    assert out.returncode == 0, out.stderr
    rec = _parse_json_line(out.stdout)
    assert REQUIRED_KEYS <= set(rec)
    assert rec["n_gpus"] == 1 and rec["steps"] == 1
    assert rec["value"] > 0 and rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"


@pytest.mark.timeout(300)
def test_bench_gloo_world2_contract():
    """The exact launch shape the driver uses for N>1, on gloo."""
    port = socket.socket()
    port.bind(("127.0.0.1", 0))
    portno = port.getsockname()[1]
    port.close()
    env_base = dict(os.environ, MASTER_ADDR="127.0.0.1",
                    MASTER_PORT=str(portno), WORLD_SIZE="2")
    procs = []
    for rank in range(2):
        env = dict(env_base, RANK=str(rank), LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--gpus", "2"] + TINY,
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True, cwd=str(REPO)))
    outs = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=240)
        assert p.returncode == 0, f"bench rank failed:\n{stderr}"
        outs.append(stdout)
    # exactly one rank (rank 0) prints the JSON line
    json_lines = [ln for o in outs for ln in o.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1
    rec = json.loads(json_lines[0])
    assert REQUIRED_KEYS <= set(rec)
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "task-dp2"
