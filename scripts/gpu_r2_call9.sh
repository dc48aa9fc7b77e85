#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH

timeout 600 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_golden.py -q > gpurun_out/r2c9_tests.log 2>&1
echo "pytest exit: $?" >> gpurun_out/r2c9_tests.log
tail -2 gpurun_out/r2c9_tests.log

timeout 600 python examples/benchmark/kernel_bench.py > gpurun_out/r2c9_kernelbench.log 2>&1
echo "exit: $?" >> gpurun_out/r2c9_kernelbench.log

timeout 300 python bench.py --steps 40 --warmup 15 --algorithm bytegrad > gpurun_out/r2c9_bytegrad.log 2>&1
echo "exit: $?" >> gpurun_out/r2c9_bytegrad.log

du -sh gpurun_out
tail -28 gpurun_out/r2c9_kernelbench.log
grep -h "^{" gpurun_out/r2c9_bytegrad.log
