#!/bin/bash
# SQ stall decomposition of the f32 zring (same recipe as pmc_bf16.sh).
repo=$(pwd)
cd /tmp && export TMPDIR=/tmp && cd "$repo"
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
  SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_LDS \
  SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
  --output-format csv -d gpurun_out/pmc_f32 -- \
  python tools/zring_ab.py > gpurun_out/pmc_f32.log 2>&1
echo pmc=$?
python - << 'PYEOF'
import csv, glob, collections
for f in glob.glob('gpurun_out/pmc_f32/**/*counter_collection.csv', recursive=True):
    for kern in ('zring_pl<28','zring_pl<36'):
        agg = collections.defaultdict(float); cnt = collections.defaultdict(int)
        with open(f) as fh:
            for row in csv.DictReader(fh):
                if 'zring_pl' in row.get('Kernel_Name','') and kern.split('<')[1].rstrip('>') in row['Kernel_Name'].split('<')[1][:3]:
                    agg[row['Counter_Name']] += float(row['Counter_Value']); cnt[row['Counter_Name']] += 1
        if agg:
            w = agg['SQ_WAVE_CYCLES']/cnt['SQ_WAVE_CYCLES']
            print(kern, {k: round((agg[k]/cnt[k])/w,3) for k in agg if k != 'SQ_WAVE_CYCLES'})
PYEOF
