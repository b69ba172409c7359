"""Guard the driver's bench.py contract: flags, JSON schema, torchrun path.

The round driver launches bench.py directly (and through torch.distributed.run
for N>1); these tests execute those exact launch shapes on CPU with a tiny
config so a contract regression is caught before a GPU round-end run.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")

REQUIRED_FIELDS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                   "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                   "dtype", "data", "config"}


def _last_json_line(out):
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


def test_bench_single_process_json_schema():
    r = subprocess.run(
        [sys.executable, BENCH, "--steps", "1", "--warmup", "0",
         "--nstack", "1", "--input", "128", "--batch", "1"],
        capture_output=True, text=True, timeout=420, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _last_json_line(r.stdout)
    assert REQUIRED_FIELDS.issubset(j.keys()), j.keys()
    # the DEFAULT invocation (what the driver runs) emits the inference FPS
    # headline with vs_baseline filled
    assert j["metric"] == "fps_512_infer"
    assert j["vs_baseline"] is not None
    assert j["n_gpus"] == 1 and j["steps"] == 1
    assert j["data"] == "synthetic"
    assert isinstance(j["config"], dict) and "global_batch" in j["config"]
    assert j["value"] > 0 and j["ms_per_step"] > 0


def test_bench_infer_mode_reports_vs_baseline():
    r = subprocess.run(
        [sys.executable, BENCH, "--mode", "infer", "--steps", "1",
         "--warmup", "0", "--nstack", "1", "--input", "128", "--batch", "1"],
        capture_output=True, text=True, timeout=420, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _last_json_line(r.stdout)
    assert j["metric"] == "fps_512_infer"
    assert j["vs_baseline"] is not None  # ratio vs the 38.5 FPS headline


@pytest.mark.timeout(600)
def test_bench_torchrun_two_ranks_cpu():
    """The driver's N>1 launch shape over gloo: one JSON line from rank 0,
    whole-job aggregate value."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", BENCH, "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--nstack", "1", "--input", "128", "--batch", "1"],
        capture_output=True, text=True, timeout=540, cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    j = _last_json_line(r.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
    assert j["config"]["global_batch"] == 2  # batch 1 per rank, aggregated


@pytest.mark.timeout(900)
def test_bench_torchrun_four_ranks_cpu():
    """The driver's N=4 launch shape (SCALE run) over gloo."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29527", BENCH, "--gpus", "4", "--steps", "1",
         "--warmup", "0", "--nstack", "1", "--input", "128", "--batch", "1",
         "--mode", "train"],
        capture_output=True, text=True, timeout=840, cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    j = _last_json_line(r.stdout)
    assert j["n_gpus"] == 4
    assert j["config"]["global_batch"] == 4
