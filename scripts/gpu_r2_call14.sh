#!/bin/bash
set -x
cd /root/repo
rm -rf gpurun_out && mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d /root/repo/gpurun_out/pmc -o kb --output-format csv -- python /root/repo/scripts/pmc_probe.py > /root/repo/gpurun_out/r2c14_pmc.log 2>&1
echo "pmc exit: $?" >> /root/repo/gpurun_out/r2c14_pmc.log
find /root/repo/gpurun_out/pmc -type f -size +8M -delete 2>/dev/null
du -sh /root/repo/gpurun_out; ls -la /root/repo/gpurun_out/pmc 2>/dev/null
tail -3 /root/repo/gpurun_out/r2c14_pmc.log
