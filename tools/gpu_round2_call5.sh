#!/bin/bash
# GPU call 5 (round 2): validate copy-path optimizations on hardware
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q > gpurun_out/gputests5.log 2>&1
echo "pytest rc=$?" >> gpurun_out/gputests5.log
CRO_BENCH_PHASES=gpurun_out/phases5.json timeout 420 \
  python bench.py --steps 100 --warmup 10 > gpurun_out/bench5.log 2>&1
echo "bench rc=$?" >> gpurun_out/bench5.log
tail -3 gpurun_out/gputests5.log
grep '^{"metric"' gpurun_out/bench5.log
cat gpurun_out/phases5.json
