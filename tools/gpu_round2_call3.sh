#!/bin/bash
# GPU call 3 (round 2): endurance bench + phase breakdown + PMC counters
set -x
mkdir -p gpurun_out
# 300-cycle endurance with phase breakdown
CRO_BENCH_PHASES=gpurun_out/phases300.json timeout 600 \
  python bench.py --steps 300 --warmup 10 --skip-extras > gpurun_out/bench300.log 2>&1
echo "bench300 rc=$?" >> gpurun_out/bench300.log
# PMC counters for the probe kernels (counters-only run: --pmc may not be
# combined with trace domains on this pool)
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES -d /root/repo/gpurun_out/pmc -- \
  python -c "from cro_amd.nodeops.probe import run_probe; print(run_probe(0))" \
  > /root/repo/gpurun_out/pmc_probe.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_probe.log
cd /root/repo
tail -2 gpurun_out/bench300.log
cat gpurun_out/phases300.json 2>/dev/null | head -30
