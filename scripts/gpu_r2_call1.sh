#!/bin/bash
# Round-2 GPU call 1: validate r2 changes + same-HW baselines + conv wrw probe
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "=== pytest -m gpu ===" > gpurun_out/r2c1_tests.log
timeout 900 python -m pytest tests -m gpu -x -q >> gpurun_out/r2c1_tests.log 2>&1
echo "pytest exit: $?" >> gpurun_out/r2c1_tests.log

echo "=== bench bagua ===" > gpurun_out/r2c1_bench.log
timeout 300 python bench.py --steps 40 --warmup 15 >> gpurun_out/r2c1_bench.log 2>&1
echo "exit: $?" >> gpurun_out/r2c1_bench.log
echo "=== bench torch-ddp ===" >> gpurun_out/r2c1_bench.log
timeout 300 python bench.py --steps 40 --warmup 15 --backend torch-ddp >> gpurun_out/r2c1_bench.log 2>&1
echo "exit: $?" >> gpurun_out/r2c1_bench.log
echo "=== bench none ===" >> gpurun_out/r2c1_bench.log
timeout 300 python bench.py --steps 40 --warmup 15 --backend none >> gpurun_out/r2c1_bench.log 2>&1
echo "exit: $?" >> gpurun_out/r2c1_bench.log

# conv wrw probe: long warmup then profiled steady window; NORMAL find mode
# with a persistent user DB so find results stick
echo "=== conv probe ===" > gpurun_out/r2c1_conv.log
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_r2c1 -o conv_probe -- python /root/repo/bench.py --steps 60 --warmup 60 >> /root/repo/gpurun_out/r2c1_conv.log 2>&1
echo "exit: $?" >> /root/repo/gpurun_out/r2c1_conv.log
# keep only the stats csv (trace files are huge)
find /root/repo/gpurun_out/prof_r2c1 -name "*kernel_trace*" -delete 2>/dev/null
true
