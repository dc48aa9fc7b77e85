#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd /tmp && export TMPDIR=/tmp

# discover available PMC counters on gfx950
timeout 120 rocprofv3 --list-avail > /root/repo/gpurun_out/r2c13_counters.txt 2>&1
grep -iE "FETCH_SIZE|WRITE_SIZE|TCC_EA|SQ_WAVES|VALUUTIL|MFMA" /root/repo/gpurun_out/r2c13_counters.txt | head -30

# PMC-only run (never combined with trace flags) over the kernel bench
timeout 600 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d /root/repo/gpurun_out/pmc -o kb -- python /root/repo/examples/benchmark/kernel_bench.py > /root/repo/gpurun_out/r2c13_pmc.log 2>&1
echo "pmc exit: $?" >> /root/repo/gpurun_out/r2c13_pmc.log
find /root/repo/gpurun_out/pmc -type f -size +8M -delete 2>/dev/null
du -sh /root/repo/gpurun_out
ls -la /root/repo/gpurun_out/pmc 2>/dev/null | head
