#!/bin/bash
cd /tmp && export TMPDIR=/tmp
cat > /tmp/wg_loop.py <<'PY'
import sys; sys.path.insert(0, "/root/repo")
import torch
from quintnet_amd import _C
dy = torch.randn(16384, 3072, device="cuda", dtype=torch.bfloat16)
x = torch.randn(16384, 768, device="cuda", dtype=torch.bfloat16)
for _ in range(20): _C.wgrad_tn(dy, x)
torch.cuda.synchronize()
PY
for counters in "SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_INSTS_LDS,SQ_LDS_BANK_CONFLICT" "SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_BUSY_CYCLES,SQ_VALU_MFMA_BUSY_CYCLES"; do
  rm -rf /tmp/wgpmc; QN_WGRAD_SPLITS=8 rocprofv3 --pmc $counters -d /tmp/wgpmc -o wg -- python /tmp/wg_loop.py >/dev/null 2>&1
  python3 - <<PY
import glob, csv, collections
files = glob.glob("/tmp/wgpmc/**/*.csv", recursive=True)
agg = collections.defaultdict(float); dur=0
for fn in files:
    for row in csv.DictReader(open(fn)):
        if "wgrad_tn" in row.get("Kernel_Name",""):
            agg[row["Counter_Name"]] += float(row["Counter_Value"])
for k,v in sorted(agg.items()):
    print(f"  {k}: {v:.3e}")
PY
done
