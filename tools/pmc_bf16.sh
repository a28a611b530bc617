#!/bin/bash
# SQ stall decomposition of the bf16 zring conv (guide: MI355X_MICROARCH
# "rocprofv3 PMC slots"): one pass, 8 SQ slots, no trace domains.
repo=$(pwd)
cd /tmp && export TMPDIR=/tmp && cd "$repo"
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
  SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_LDS \
  SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
  --output-format csv -d gpurun_out/pmc_bf16 -- \
  python tools/conv_probe.py --bf16 > gpurun_out/pmc_bf16.log 2>&1
echo pmc=$?
find gpurun_out/pmc_bf16 -name "*.csv" | head
python - << 'PYEOF'
import csv, glob, collections
for f in glob.glob('gpurun_out/pmc_bf16/**/*counter*.csv', recursive=True) or glob.glob('gpurun_out/pmc_bf16/**/*.csv', recursive=True):
    agg = collections.defaultdict(float); cnt = collections.defaultdict(int)
    with open(f) as fh:
        for row in csv.DictReader(fh):
            kn = row.get('Kernel_Name','')
            if 'zring_bf16' in kn:
                agg[row['Counter_Name']] += float(row['Counter_Value']); cnt[row['Counter_Name']] += 1
    if agg:
        print(f)
        for k in sorted(agg): print(f'  {k}: {agg[k]/max(cnt[k],1):.3e} (n={cnt[k]})')
PYEOF
