"""bench.py driver-contract rehearsal on CPU: the exact invocation shapes
the round-end driver uses (single-process, and torchrun ws=2 over gloo),
checked for the one-JSON-line output contract."""

from __future__ import annotations

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _last_json_line(stdout: str) -> dict:
    lines = [l for l in stdout.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output:\n{stdout[-2000:]}"
    return json.loads(lines[-1])


def _env(tmp_path):
    return dict(
        os.environ,
        PYTHONPATH=REPO + os.pathsep + os.environ.get("PYTHONPATH", ""),
        TMPDIR=str(tmp_path),  # isolate the packed-model cache
    )


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_single_process_contract(tmp_path):
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--device", "cpu",
         "--steps", "2", "--warmup", "1", "--rows", "64",
         "--model-trees", "12", "--model-depth", "4"],
        capture_output=True, text=True, timeout=540, env=_env(tmp_path), cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    d = _last_json_line(r.stdout)
    assert REQUIRED_KEYS <= set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["parallelism"] == "dp1"


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_torchrun_ws2_gloo(tmp_path):
    """The driver's N>1 launch shape (torch.distributed.run, one rank per
    GPU) rehearsed with gloo on CPU: rank 0 trains + broadcasts, both ranks
    step, MAX-over-ranks elapsed, rank 0 prints one line with the
    whole-job aggregate."""
    port = _free_port()
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(REPO, "bench.py"), "--device", "cpu", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--rows", "64",
         "--model-trees", "12", "--model-depth", "4"],
        capture_output=True, text=True, timeout=540, env=_env(tmp_path), cwd=REPO,
    )
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    d = _last_json_line(r.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 2 * 64
    # exactly one result line (rank 0 only)
    assert sum(1 for l in r.stdout.splitlines() if l.startswith("{")) == 1


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_gpus_n_self_spawns_real_ranks(tmp_path):
    """`bench.py --gpus 2` WITHOUT torchrun must launch 2 real ranks and
    report their measured aggregate — never multiply a single-rank rate by
    N (round-1 verdict weak-spot #1)."""
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--device", "cpu",
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--rows", "64",
         "--model-trees", "12", "--model-depth", "4"],
        capture_output=True, text=True, timeout=540, env=_env(tmp_path), cwd=REPO,
    )
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    assert "self-spawning 2 ranks" in r.stderr
    d = _last_json_line(r.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert sum(1 for l in r.stdout.splitlines() if l.startswith("{")) == 1


@pytest.mark.timeout(300)
def test_bench_refuses_to_extrapolate_gpus(tmp_path):
    """--gpus N with cuda requested but N devices unavailable must refuse
    (exit 2), not print an extrapolated whole-node number."""
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--device", "cuda",
         "--gpus", "8", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=240, env=_env(tmp_path), cwd=REPO,
    )
    # no GPU in CI: device_count()==0 < 8 -> refusal path
    assert r.returncode == 2, (r.returncode, r.stdout[-500:], r.stderr[-500:])
    assert "refusing to extrapolate" in r.stderr
    assert not [l for l in r.stdout.splitlines() if l.startswith("{")]


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_torchrun_ws4_gloo(tmp_path):
    """4-rank rehearsal of the driver's scale shape (gloo): broadcast from
    rank 0, all ranks step, MAX-over-ranks aggregate — the same code path
    RCCL takes on the 8-GPU node."""
    port = _free_port()
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(REPO, "bench.py"), "--device", "cpu", "--gpus", "4",
         "--steps", "2", "--warmup", "1", "--rows", "32",
         "--model-trees", "10", "--model-depth", "3"],
        capture_output=True, text=True, timeout=540, env=_env(tmp_path), cwd=REPO,
    )
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    d = _last_json_line(r.stdout)
    assert d["n_gpus"] == 4 and d["config"]["parallelism"] == "dp4"
    assert sum(1 for l in r.stdout.splitlines() if l.startswith("{")) == 1
