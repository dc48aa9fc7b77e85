#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH

timeout 600 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_golden.py -q > gpurun_out/r2c4_tests.log 2>&1
echo "pytest exit: $?" >> gpurun_out/r2c4_tests.log
tail -2 gpurun_out/r2c4_tests.log

timeout 900 python scripts/exec_overhead_bench.py > gpurun_out/r2c4_exec.log 2>&1
echo "exit: $?" >> gpurun_out/r2c4_exec.log

timeout 600 python scripts/autotune_gpu_probe.py > gpurun_out/r2c4_autotune.log 2>&1
echo "exit: $?" >> gpurun_out/r2c4_autotune.log

du -sh gpurun_out
grep -h "^{" gpurun_out/r2c4_exec.log
grep -h "SUMMARY" gpurun_out/r2c4_autotune.log
