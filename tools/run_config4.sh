#!/bin/bash
# BASELINE config 4 at full size through the real CLI, plus a final
# rocprofv3 stats capture of the default bench (run on the GPU box).
mkdir -p gpurun_out/c4log3
timeout 400 python -m chunkflow_amd.flow \
  generate-tasks --roi-size 512 1024 1024 --chunk-size 256 1024 1024 \
  create-chunk --dtype uint8 --pattern sin \
  normalize-intensity \
  inference --convnet-model examples/nets/rsunet.py -s 20 256 256 \
    --output-patch-overlap 4 64 64 --framework pytorch --batch-size 12 \
    --num-output-channels 3 --mask-output-chunk \
  crop-margin \
  connected-components -t 0.0 -c 6 \
  save-log -o gpurun_out/c4log3 > gpurun_out/c4run3.log 2>&1
echo chain=$?
cat gpurun_out/c4log3/*.json 2>/dev/null
repo=$(pwd)
cd /tmp && export TMPDIR=/tmp && cd "$repo"
timeout 500 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_final -- \
  python bench.py --steps 1 --warmup 1 > gpurun_out/bench_prof_final.log 2>&1
echo prof=$?
tail -1 gpurun_out/bench_prof_final.log
