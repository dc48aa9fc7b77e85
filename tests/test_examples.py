"""Examples double as integration tests (reference pattern:
examples/communication_primitives/main.py used by CI)."""

import subprocess
import sys

from tests.internal.multi_process import find_free_port


def _run(cmd, timeout=240):
    out = subprocess.run(cmd, capture_output=True, text=True,
                         timeout=timeout)
    assert out.returncode == 0, out.stderr[-3000:]
    return out.stdout


def test_communication_primitives_example():
    out = _run([sys.executable, "-m", "bagua_amd.distributed.run",
                "--standalone", "--nnodes=1", "--nproc-per-node=2",
                "--local-addr", "127.0.0.1",
                "examples/communication_primitives/main.py"])
    assert "all communication primitives verified" in out


def test_mnist_example_two_ranks():
    out = _run([sys.executable, "-m", "bagua_amd.distributed.launch",
                "--nproc_per_node", "2", "--master_port",
                str(find_free_port()),
                "examples/mnist/main.py", "--epochs", "1",
                "--batches-per-epoch", "5", "--batch-size", "16"])
    assert "done" in out


def test_moe_example_two_ranks():
    out = _run([sys.executable, "-m", "bagua_amd.distributed.launch",
                "--nproc_per_node", "2", "--master_port",
                str(find_free_port()),
                "examples/moe/main.py", "--steps", "5",
                "--batch-size", "16"])
    assert "checkpoint saved+restored" in out


def test_imagenet_example_two_ranks():
    out = _run([sys.executable, "-m", "bagua_amd.distributed.launch",
                "--nproc_per_node", "2", "--master_port",
                str(find_free_port()),
                "examples/imagenet/main.py", "--arch", "resnet50",
                "--epochs", "1", "--batches-per-epoch", "3",
                "--batch-size", "4"])
    assert "img/s (whole job)" in out


def test_squad_example_two_ranks():
    out = _run([sys.executable, "-m", "bagua_amd.distributed.launch",
                "--nproc_per_node", "2", "--master_port",
                str(find_free_port()),
                "examples/squad/main.py", "--steps", "3",
                "--batch-size", "2", "--seq-len", "64"])
    assert "tokens/s (whole job)" in out
