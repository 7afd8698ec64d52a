#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
CTRS1="SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_WAVE_CYCLES"
CTRS2="TCP_UTCL1_TRANSLATION_MISS TCC_REQ TCC_MISS TCC_EA0_RDREQ_DRAM_CREDIT_STALL"
timeout 300 rocprofv3 --pmc $CTRS1 --output-format csv -d gpurun_out/pmc_attn1 -o a \
  -- python benchmarks/bench_kernels.py attn_decode > gpurun_out/attn_pmc1.log 2>&1
echo "pmc1 rc=$?"
timeout 300 rocprofv3 --pmc $CTRS2 --output-format csv -d gpurun_out/pmc_attn2 -o b \
  -- python benchmarks/bench_kernels.py attn_decode > gpurun_out/attn_pmc2.log 2>&1
echo "pmc2 rc=$?"
grep "ns=" gpurun_out/attn_pmc1.log | head -20
timeout 120 python - > gpurun_out/attn_pmc_summary.txt 2>&1 <<'PYEOF'
import csv, glob, collections
agg = collections.defaultdict(lambda: collections.defaultdict(float))
n = collections.Counter()
for f in glob.glob("gpurun_out/pmc_attn*/**/*counter*.csv", recursive=True):
    for row in csv.DictReader(open(f)):
        kn = (row.get("Kernel_Name") or "").split("(")[0][:70]
        cn = row.get("Counter_Name"); cv = row.get("Counter_Value")
        if not cn or "attn" not in kn: continue
        agg[kn][cn] += float(cv); n[kn] += 1
for k, c in sorted(agg.items()):
    print(f"{k} (n={n[k]})")
    for cn, cv in sorted(c.items()): print(f"    {cn} = {cv:.4e}")
    wc = c.get("SQ_WAVE_CYCLES"); wa = c.get("SQ_WAIT_ANY"); wi = c.get("SQ_WAIT_INST_ANY"); ac = c.get("SQ_ACTIVE_INST_ANY")
    if wc:
        print(f"    -> WAIT_ANY {100*wa/wc:.1f}%  WAIT_INST {100*wi/wc:.1f}%  ACTIVE {100*ac/wc:.1f}%")
    tr, tm = c.get("TCC_REQ"), c.get("TCC_MISS")
    if tr: print(f"    -> L2 miss {100*tm/tr:.1f}%")
PYEOF
cat gpurun_out/attn_pmc_summary.txt
find gpurun_out/pmc_attn1 gpurun_out/pmc_attn2 \( -name '*.csv' -size +8M -o -name '*.db' \) -delete 2>/dev/null
echo DONE
