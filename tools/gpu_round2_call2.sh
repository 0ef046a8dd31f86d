#!/bin/bash
# GPU call 2 (round 2): full GPU tests + smoke + bench with all secondary
# configs after the protocol/auth/lease/backoff changes
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q > gpurun_out/gputests2.log 2>&1
echo "pytest rc=$?" >> gpurun_out/gputests2.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke2.log 2>&1
echo "smoke rc=$?" >> gpurun_out/smoke2.log
timeout 420 python bench.py --steps 20 --warmup 5 > gpurun_out/bench2.log 2>&1
echo "bench rc=$?" >> gpurun_out/bench2.log
tail -3 gpurun_out/gputests2.log
tail -2 gpurun_out/smoke2.log
tail -2 gpurun_out/bench2.log
