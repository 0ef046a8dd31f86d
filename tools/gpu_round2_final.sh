#!/bin/bash
# Final round-2 validation: full GPU suite + smoke + both bench modes
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q > gpurun_out/gputests_final.log 2>&1
echo "pytest rc=$?" >> gpurun_out/gputests_final.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke_final.log 2>&1
echo "smoke rc=$?" >> gpurun_out/smoke_final.log
timeout 420 python bench.py --steps 50 --warmup 10 > gpurun_out/bench_final.log 2>&1
echo "bench rc=$?" >> gpurun_out/bench_final.log
timeout 300 python bench.py --steps 30 --warmup 5 --mode DEVICE_PLUGIN --skip-extras --force-detach > gpurun_out/bench_final_dp.log 2>&1
echo "bench-dp rc=$?" >> gpurun_out/bench_final_dp.log
tail -3 gpurun_out/gputests_final.log
tail -2 gpurun_out/smoke_final.log
grep '^{"metric"' gpurun_out/bench_final.log gpurun_out/bench_final_dp.log
