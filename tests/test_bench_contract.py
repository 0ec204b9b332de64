"""bench.py driver contract: single-process default run and the exact
torchrun multi-rank launch shape the round-end driver uses (CPU/gloo here;
the same code path binds NCCL/RCCL per-GPU on an MI355X node)."""
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def last_json_line(text):
    for line in reversed(text.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{text[-2000:]}")


def test_bench_single_process_defaults():
    r = subprocess.run(
        [sys.executable, BENCH, "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    out = last_json_line(r.stdout)
    for k in REQUIRED:
        assert k in out, f"missing field {k}"
    assert out["n_gpus"] == 1 and out["steps"] == 2
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["higher_is_better"] is True and out["scaling"] == "weak"
    assert out["vs_baseline"] is None


def test_bench_torchrun_two_ranks():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), BENCH,
         "--gpus", "2", "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    out = last_json_line(r.stdout)
    assert out["n_gpus"] == 2
    # whole-job aggregate: 2 ranks of tiny-cpu x 4 users x 2 steps
    assert out["value"] > 0
    # exactly ONE json line (rank 0 only prints)
    n_json = sum(1 for l in r.stdout.splitlines()
                 if l.strip().startswith("{") and '"metric"' in l)
    assert n_json == 1, r.stdout


def test_bench_stack_mode():
    """--stack measures through the full dispatcher + UDS worker path and
    reports the dispatcher's own queue-wait percentiles."""
    r = subprocess.run(
        [sys.executable, BENCH, "--stack", "--gpus", "1",
         "--steps", "36", "--warmup", "3"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    out = last_json_line(r.stdout)
    for k in REQUIRED:
        assert k in out, f"missing field {k}"
    assert out["value"] > 0
    cfg = out["config"]
    assert "full-stack" in cfg["mode"]
    assert cfg["p50_queue_wait_ms"] is not None
    assert cfg["requests_processed"] > 0
