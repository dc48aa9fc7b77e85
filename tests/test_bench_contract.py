"""bench.py contract: runs under torch.distributed.run exactly as the
driver launches it, emits one valid JSON line from rank 0."""

import json
import subprocess
import sys

from tests.internal.multi_process import find_free_port


def test_bench_distributed_cpu():
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--standalone", "--local-addr",
         "127.0.0.1", "bench.py", "--gpus", "2", "--steps", "3",
         "--warmup", "1", "--model", "mnist", "--batch-size", "8",
         "--dtype", "fp32"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in rec, "missing %s" % key
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 3
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0


def test_bench_single_process_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "mnist", "--batch-size", "4", "--dtype", "fp32"],
        capture_output=True, text=True, timeout=240,
        env={"PATH": "/usr/bin:/bin:/usr/local/bin",
             "PYTHONPATH": ".",
             "MASTER_PORT": str(find_free_port()), "HOME": "/root"})
    assert out.returncode == 0, out.stderr[-3000:]
    rec = json.loads([ln for ln in out.stdout.splitlines()
                      if ln.startswith("{")][0])
    assert rec["n_gpus"] == 1
