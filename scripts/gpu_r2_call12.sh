#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH

timeout 1200 python -m pytest tests -m gpu -q > gpurun_out/r2c12_tests.log 2>&1
echo "pytest exit: $?" >> gpurun_out/r2c12_tests.log
tail -2 gpurun_out/r2c12_tests.log

timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r2c12_smoke.log 2>&1
echo "exit: $?" >> gpurun_out/r2c12_smoke.log
tail -2 gpurun_out/r2c12_smoke.log

timeout 300 python bench.py --steps 40 --warmup 15 > gpurun_out/r2c12_bench.log 2>&1
echo "exit: $?" >> gpurun_out/r2c12_bench.log

du -sh gpurun_out
grep -h "^{" gpurun_out/r2c12_bench.log
