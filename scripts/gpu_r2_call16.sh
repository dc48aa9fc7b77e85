#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH

timeout 1200 python -m pytest tests -m gpu -q > gpurun_out/r2c16_tests.log 2>&1
echo "pytest exit: $?" >> gpurun_out/r2c16_tests.log
tail -2 gpurun_out/r2c16_tests.log

timeout 1500 bash scripts/ci_determinism.sh 1 > gpurun_out/r2c16_det.log 2>&1
echo "det exit: $?" >> gpurun_out/r2c16_det.log
grep -hE "determinism|FAIL" gpurun_out/r2c16_det.log | tail -7

du -sh gpurun_out
