#!/bin/bash
# GPU call 4 (round 2): node-path microbench + PMC retry
set -x
mkdir -p gpurun_out
timeout 300 python tools/microbench_nodepath.py --iters 50 > gpurun_out/microbench.json 2>gpurun_out/microbench.err
echo "microbench rc=$?" >> gpurun_out/microbench.err
cd /tmp && export TMPDIR=/tmp PYTHONPATH=/root/repo
timeout 300 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES -d /root/repo/gpurun_out/pmc -- \
  python -c "from cro_amd.nodeops.probe import run_probe; print(run_probe(0))" \
  > /root/repo/gpurun_out/pmc_probe.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_probe.log
cd /root/repo
head -40 gpurun_out/microbench.json
tail -2 gpurun_out/pmc_probe.log
