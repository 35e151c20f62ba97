"""bench.py driver contract: ONE parseable JSON line with the required
fields, single-process and through a real 2-rank torchrun gloo launch
(the CPU analog of the driver's BENCH/SCALE runs)."""
import json
import subprocess
import sys

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _last_json(stdout: str) -> dict:
    lines = [ln for ln in stdout.splitlines() if ln.startswith("{")]
    assert lines, f"no JSON line in output:\n{stdout[-500:]}"
    return json.loads(lines[-1])


def test_bench_single_process_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "256", "--no-hip-graph"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    d = _last_json(r.stdout)
    assert REQUIRED <= set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["config"]["model"] == "dlrm"
    assert abs(d["value"] - d["config"]["global_batch"] * 1000.0
               / d["ms_per_step"]) / d["value"] < 1e-6


def test_bench_two_rank_gloo_contract():
    """torchrun --nproc-per-node 2: exactly rank 0 prints the JSON,
    n_gpus=2, whole-job value (2x per-rank batch), dp2+ep2 label."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29574", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "128"],
        capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, (r.stdout[-400:], r.stderr[-800:])
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, "exactly one rank must print the JSON"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["global_batch"] == 256  # whole node, not per rank
    assert d["config"]["parallelism"] == "dp2+ep2"
    assert d["value"] > 0


def test_bench_force_dist_single_process():
    """--force-dist at world=1 on CPU: the sharded+collective code path
    (gloo self-exchange) behind the same JSON contract."""
    r = subprocess.run(
        [sys.executable, "bench.py", "--force-dist", "--steps", "2",
         "--warmup", "1", "--batch", "128"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    d = _last_json(r.stdout)
    assert d["config"]["parallelism"] == "dp1+ep1"
    assert d["value"] > 0


def test_graft_entry_surface():
    import __graft_entry__
    assert callable(__graft_entry__.build)
    assert callable(__graft_entry__.smoke)
