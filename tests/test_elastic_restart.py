"""Elastic recovery: torchelastic gang-restart + checkpoint resume
(reference behavior: distributed/run.py restart semantics + elastic
example self-checkpointing). Injects a worker crash after the first
epoch's checkpoint and asserts the restarted group resumes from it."""

import subprocess
import sys


def test_worker_crash_restart_resumes(tmp_path):
    ckpt = tmp_path / "ckpt"
    crash_flag = tmp_path / "crashed_once"
    script = tmp_path / "train.py"
    script.write_text(f"""
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, {str(repr(__import__('os').getcwd()))})

os.environ["BAGUA_PG_TIMEOUT_S"] = "20"  # fail fast after gang crash

import bagua_amd
from bagua_amd.checkpoint import load_checkpoint, save_checkpoint
from bagua_amd.models import MnistNet
from bagua_amd.parallel.algorithms.gradient_allreduce import (
    GradientAllReduceAlgorithm,
)

bagua_amd.init_process_group()
torch.manual_seed(13)
model = MnistNet()
optimizer = torch.optim.SGD(model.parameters(), lr=0.01)

start_epoch = 0
if os.path.isdir({str(repr(str(ckpt)))}):
    start_epoch = load_checkpoint({str(repr(str(ckpt)))}, model, optimizer)
    if start_epoch:
        print("RESUMED_FROM", start_epoch, flush=True)

ddp = bagua_amd.DistributedDataParallel(
    model, optimizers=[optimizer], algorithm=GradientAllReduceAlgorithm())

for epoch in range(start_epoch, 3):
    for b in range(2):
        torch.manual_seed(epoch * 100 + b + bagua_amd.get_rank())
        data = torch.randn(8, 1, 28, 28)
        target = torch.randint(0, 10, (8,))
        optimizer.zero_grad()
        F.nll_loss(ddp(data), target).backward()
        optimizer.step()
    save_checkpoint(epoch + 1, {str(repr(str(ckpt)))}, model, optimizer)
    # injected fault: rank 0 dies after epoch 1's checkpoint, once
    if epoch == 0 and bagua_amd.get_rank() == 0 \\
            and not os.path.exists({str(repr(str(crash_flag)))}):
        open({str(repr(str(crash_flag)))}, "w").write("x")
        print("INJECTED_CRASH", flush=True)
        os._exit(17)
print("FINISHED", flush=True)
""")
    out = subprocess.run(
        [sys.executable, "-m", "bagua_amd.distributed.run",
         "--standalone", "--nnodes=1", "--nproc-per-node=2",
         "--local-addr", "127.0.0.1", "--max-restarts", "5",
         str(script)],
        capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-3000:]
    assert "INJECTED_CRASH" in out.stdout
    assert "RESUMED_FROM 1" in out.stdout, out.stdout[-2000:]
    assert "FINISHED" in out.stdout
