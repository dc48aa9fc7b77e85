#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH

timeout 1200 python -m pytest tests -m gpu -q > gpurun_out/r2c3_tests.log 2>&1
echo "pytest exit: $?" >> gpurun_out/r2c3_tests.log
tail -3 gpurun_out/r2c3_tests.log

timeout 600 python scripts/exec_overhead_bench.py > gpurun_out/r2c3_exec.log 2>&1
echo "exit: $?" >> gpurun_out/r2c3_exec.log

timeout 600 python scripts/autotune_gpu_probe.py > gpurun_out/r2c3_autotune.log 2>&1
echo "exit: $?" >> gpurun_out/r2c3_autotune.log

timeout 300 python -m bagua_amd.distributed.sys_perf --min-bytes $((1<<20)) --max-bytes $((1<<28)) > gpurun_out/r2c3_sysperf.log 2>&1
echo "exit: $?" >> gpurun_out/r2c3_sysperf.log

du -sh gpurun_out
grep -h "^{\|SUMMARY" gpurun_out/r2c3_exec.log gpurun_out/r2c3_autotune.log | tail -20
