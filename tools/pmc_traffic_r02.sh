#!/bin/bash
# Re-collect the blend kernel's per-launch HBM traffic on the RSUNET
# config-2 mix (the committed constant came from the
# identity mix). Two separate --pmc passes (FETCH_SIZE costs 3 TCC slots,
# WRITE_SIZE 2 — they don't fit one pass; guide §rocprofv3 PMC slots);
# gfx950 FETCH_SIZE reports HALF the bytes of a wide coalesced read.
repo=$(pwd)
cd /tmp && export TMPDIR=/tmp && cd "$repo"
EXTRA="${BENCH_ARGS:-}"
TAG="${MIX_TAG:-}"
for ctr in FETCH_SIZE WRITE_SIZE; do
  timeout 500 rocprofv3 --pmc $ctr --output-format csv \
    -d gpurun_out/pmc_traffic$TAG-$ctr -- \
    python bench.py --steps 1 --warmup 1 --no-cpu-baseline $EXTRA \
    > gpurun_out/pmc_traffic$TAG-$ctr.log 2>&1
  echo "$ctr rc=$?"
done
python - << 'PYEOF'
import csv, glob, json, collections
res = {}
for ctr in ('FETCH_SIZE', 'WRITE_SIZE'):
    agg = collections.defaultdict(float)
    cnt = collections.defaultdict(int)
    import os
    mixtag = os.environ.get('MIX_TAG', '')
    for f in glob.glob(f'gpurun_out/pmc_traffic{mixtag}-{ctr}/**/*.csv',
                       recursive=True):
        with open(f) as fh:
            for row in csv.DictReader(fh):
                kn = row.get('Kernel_Name', '')
                if row.get('Counter_Name') != ctr:
                    continue
                for tag in ('k_blend_batch', 'k_extract', 'k_maskmul',
                            'k_conv3_zring_pl', 'k_upconv2',
                            'k_conv155'):
                    if tag in kn:
                        agg[tag] += float(row['Counter_Value'])
                        cnt[tag] += 1
    res[ctr] = {k: {'kb_total': agg[k], 'launches': cnt[k],
                    'kb_avg': agg[k] / max(cnt[k], 1)} for k in agg}
import os
with open('gpurun_out/pmc_traffic_r02%s.json' % os.environ.get('MIX_TAG', ''),
          'w') as f:
    json.dump(res, f, indent=1)
print(json.dumps(res, indent=1))
PYEOF
