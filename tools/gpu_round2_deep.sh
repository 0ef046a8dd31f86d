#!/bin/bash
set -x
mkdir -p gpurun_out
timeout 1000 python tools/chaos_soak.py --minutes 15 --seed 13 > gpurun_out/chaos_r02b.json 2>gpurun_out/chaos_r02b.err
echo "chaos rc=$?" >> gpurun_out/chaos_r02b.err
timeout 300 python bench.py --steps 1000 --warmup 20 --skip-extras > gpurun_out/bench1000.log 2>&1
echo "bench rc=$?" >> gpurun_out/bench1000.log
tail -1 gpurun_out/chaos_r02b.json
grep '^{"metric"' gpurun_out/bench1000.log
