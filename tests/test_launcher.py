"""Launcher smoke tests (CPU, gloo)."""

import subprocess
import sys

from tests.internal.multi_process import find_free_port


def test_legacy_launch_two_procs(tmp_path):
    script = tmp_path / "train.py"
    script.write_text(
        "import bagua_amd, torch, os\n"
        "bagua_amd.init_process_group()\n"
        "t = torch.ones(4)\n"
        "bagua_amd.allreduce_inplace(t, op=bagua_amd.ReduceOp.SUM)\n"
        "assert t[0].item() == 2, t\n"
        "print('rank', os.environ['RANK'], 'ok')\n")
    out = subprocess.run(
        [sys.executable, "-m", "bagua_amd.distributed.launch",
         "--nproc_per_node", "2", "--master_port",
         str(find_free_port()), str(script)],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert "ok" in out.stdout


def test_elastic_run_two_procs(tmp_path):
    script = tmp_path / "train.py"
    script.write_text(
        "import bagua_amd, torch\n"
        "bagua_amd.init_process_group()\n"
        "t = torch.ones(2)\n"
        "bagua_amd.allreduce_inplace(t, op=bagua_amd.ReduceOp.SUM)\n"
        "assert t[0].item() == 2\n"
        "print('elastic ok')\n")
    out = subprocess.run(
        [sys.executable, "-m", "bagua_amd.distributed.run",
         "--standalone", "--nnodes=1", "--nproc-per-node=2",
         "--local-addr", "127.0.0.1", str(script)],
        capture_output=True, text=True, timeout=180)
    assert out.returncode == 0, out.stderr
    assert "elastic ok" in out.stdout


def test_baguarun_command_build():
    from bagua_amd.distributed.baguarun import build_remote_command, parse_args

    args = parse_args(["--host_list", "h1,h2", "--nproc_per_node", "4",
                       "train.py", "--lr", "0.1"])
    cmd = build_remote_command(args, node_rank=1, master_addr="h1")
    assert "--node_rank 1" in cmd and "--master_addr h1" in cmd
    assert "train.py --lr 0.1" in cmd


def test_sys_perf_cli_two_procs():
    out = subprocess.run(
        [sys.executable, "-m", "bagua_amd.distributed.run",
         "--standalone", "--nnodes=1", "--nproc-per-node=2",
         "--local-addr", "127.0.0.1", "-m",
         "bagua_amd.distributed.sys_perf",
         "--min-bytes", "4096", "--max-bytes", "16384",
         "--iters", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "busbw" in out.stdout
    assert '"world_size": 2' in out.stdout
