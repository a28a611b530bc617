#!/bin/bash
# rocprofv3 stats of the final config-5 bf16 bench (evidence: the bf16
# z-ring dominates and its profiled avg agrees with the clean probe).
repo=$(pwd)
cd /tmp && export TMPDIR=/tmp && cd "$repo"
timeout 500 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_c5 -- \
  python bench.py --dtype bfloat16 --patch-size 32 256 256 --batch-size 24 \
  --steps 1 --warmup 1 > gpurun_out/bench_c5_prof.log 2>&1
echo prof=$?
ls gpurun_out/prof_c5/*/ | head -3
