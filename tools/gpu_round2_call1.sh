#!/bin/bash
# GPU call 1 (round 2): GPU tests + restructured bench + rocprof stats
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q > gpurun_out/gputests.log 2>&1
echo "pytest rc=$?" >> gpurun_out/gputests.log
timeout 420 python bench.py --steps 20 --warmup 5 > gpurun_out/bench1.log 2>&1
echo "bench rc=$?" >> gpurun_out/bench1.log
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -- \
  python /root/repo/bench.py --steps 5 --warmup 2 --skip-extras \
  > /root/repo/gpurun_out/bench_prof.log 2>&1
echo "rocprof rc=$?" >> /root/repo/gpurun_out/bench_prof.log
tail -3 /root/repo/gpurun_out/gputests.log
tail -2 /root/repo/gpurun_out/bench1.log
