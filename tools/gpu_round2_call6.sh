#!/bin/bash
# GPU call 6 (round 2): chaos soak on hardware with the optimized store
set -x
mkdir -p gpurun_out
timeout 700 python tools/chaos_soak.py --minutes 8 --seed 7 > gpurun_out/chaos_r02.json 2>gpurun_out/chaos_r02.err
echo "chaos rc=$?" >> gpurun_out/chaos_r02.err
tail -1 gpurun_out/chaos_r02.json
tail -2 gpurun_out/chaos_r02.err
