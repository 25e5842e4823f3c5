#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
# extract the PMC probe from call8 script (between PYEOF markers)
sed -n '/cat > \/tmp\/pmc_probe.py/,/^PYEOF$/p' scripts/gpu_r2_call8.sh | sed '1d;$d' > /tmp/pmc_probe.py
head -5 /tmp/pmc_probe.py
timeout 600 rocprofv3 --pmc SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT \
    SQ_WAVES -d gpurun_out/pmc9 --output-format csv -o newk \
    -- python /tmp/pmc_probe.py > gpurun_out/pmc9.log 2>&1
echo "pmc rc=$?"
tail -3 gpurun_out/pmc9.log
find gpurun_out/pmc9 -name "*.csv"
echo DONE
