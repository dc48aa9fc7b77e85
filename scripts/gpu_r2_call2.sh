#!/bin/bash
# Round-2 GPU call 2: re-collect bench JSONs + conv steady-state stats
# (call 1 ran green but merge-back was dropped: >64MiB in gpurun_out)
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 300 python bench.py --steps 40 --warmup 15 > gpurun_out/b_bagua.log 2>&1
timeout 300 python bench.py --steps 40 --warmup 15 --backend torch-ddp > gpurun_out/b_ddp.log 2>&1
timeout 300 python bench.py --steps 40 --warmup 15 --backend none > gpurun_out/b_none.log 2>&1

export MIOPEN_USER_DB_PATH=/tmp/miopen_udb
mkdir -p $MIOPEN_USER_DB_PATH
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o cp -- python /root/repo/bench.py --steps 60 --warmup 60 > /root/repo/gpurun_out/b_prof.log 2>&1
# keep ONLY small stats files
find /root/repo/gpurun_out/prof -type f ! -name "*stats*" -delete
find /root/repo/gpurun_out/prof -type f -size +4M -delete
du -sh /root/repo/gpurun_out
# echo the JSON lines to stdout as a backup channel
grep -h "^{" /root/repo/gpurun_out/b_*.log
