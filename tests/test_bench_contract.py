"""Guards the driver's bench.py contract: one JSON line from rank 0 with
the required schema, on both the single-process and torchrun paths."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REQUIRED = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _last_json_line(out: str) -> dict:
    lines = [l for l in out.strip().splitlines() if l.startswith("{")]
    assert lines, out
    return json.loads(lines[-1])


def test_bench_single_process_schema():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json_line(out.stdout)
    assert REQUIRED <= set(r)
    assert r["metric"] == "zmw_per_sec" and r["n_gpus"] == 1
    assert r["value"] > 0 and r["ms_per_step"] > 0
    assert r["higher_is_better"] is True and r["scaling"] == "weak"
    assert r["config"]["parallelism"] == "dp1"
    assert r["vs_baseline"] > 0


def test_bench_torchrun_world2_gloo():
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29733", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=900, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json_line(out.stdout)
    assert r["n_gpus"] == 2 and r["config"]["parallelism"] == "dp2"
    # value is the whole-job aggregate across ranks.
    assert r["config"]["global_batch"] == 64  # 32 per rank on CPU


def test_bench_pipeline_mode_schema():
    """--pipeline emits the JSON contract with per-stage seconds."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--pipeline", "--pipeline-zmws", "4",
         "--pipeline-length", "800"],
        cwd=REPO, capture_output=True, text=True, timeout=900,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json_line(out.stdout)
    assert r["metric"] == "zmw_per_sec_pipeline"
    assert r["value"] > 0 and r["higher_is_better"] is True
    assert r["config"]["mode"] == "whole_pipeline"
    assert r["config"]["reads_written"] == 4
    assert "run_model" in r["config"]["stage_seconds"]
