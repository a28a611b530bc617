#!/bin/bash
# BASELINE config 2 through the real CLI on one MI355X: 512^3 synthetic
# chunk -> RSUNet inference -> crop -> npy. (bench.py is the measured
# version of this pipeline; this script is the drop-in usage example.)
set -e
cd "$(dirname "$0")/.."
python -m chunkflow_amd \
  create-chunk --size 512 512 512 --dtype uint8 --pattern sin \
  inference -m examples/nets/rsunet.py -s 20 256 256 \
            --output-patch-overlap 4 64 64 -f pytorch -b 12 -c 3 \
            --mask-output-chunk \
  crop-margin -m 4 64 64 4 64 64 \
  save-npy -f /tmp/affinity.npy
