#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH

timeout 1500 bash scripts/ci_determinism.sh 1 > gpurun_out/r2c6_determinism.log 2>&1
echo "exit: $?" >> gpurun_out/r2c6_determinism.log

timeout 300 python bench.py --steps 20 --warmup 8 --model bert-large --batch-size 8 > gpurun_out/r2c6_bert.log 2>&1
echo "exit: $?" >> gpurun_out/r2c6_bert.log
timeout 300 python bench.py --steps 30 --warmup 10 --model resnet50 --batch-size 64 > gpurun_out/r2c6_resnet.log 2>&1
echo "exit: $?" >> gpurun_out/r2c6_resnet.log

du -sh gpurun_out
grep -h "algorithm=" gpurun_out/r2c6_determinism.log
grep -h "^{" gpurun_out/r2c6_bert.log gpurun_out/r2c6_resnet.log
