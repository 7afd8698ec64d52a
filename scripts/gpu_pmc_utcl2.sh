#!/bin/bash
# UTCL1-TLB + L2 PMC probe for the in-context decode-GEMM parking question
# (ROUND3.md item 1): same counters on (a) the standalone GEMM micro and
# (b) the full bench.py decode step, compare gemm_skinny translation/L2
# miss rates. Counter names verified against rocprofv3 -L on gfx950.
set -x
mkdir -p gpurun_out
cd /root/repo

CTRS="TCP_UTCL1_TRANSLATION_HIT TCP_UTCL1_TRANSLATION_MISS TCC_REQ TCC_MISS"

echo "=== standalone GEMM micro ==="
timeout 300 rocprofv3 --pmc $CTRS --output-format csv \
  -d gpurun_out/pmc_gemm_micro -o m \
  -- python benchmarks/bench_kernels.py gemm > gpurun_out/pmc_gemm_micro.log 2>&1
echo "micro rc=$?"

echo "=== in-context bench.py decode ==="
timeout 300 rocprofv3 --pmc $CTRS --output-format csv \
  -d gpurun_out/pmc_bench -o b \
  -- python bench.py --gpus 1 --steps 4 --warmup 2 \
  > gpurun_out/pmc_bench.log 2>&1
echo "bench rc=$?"

timeout 120 python - > gpurun_out/pmc_utcl2_summary.txt 2>&1 <<'PYEOF'
import csv, glob, collections
def summarize(pat):
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    n = collections.Counter()
    for f in glob.glob(pat, recursive=True):
        with open(f) as fh:
            for row in csv.DictReader(fh):
                kn = row.get("Kernel_Name") or row.get("Kernel Name") or ""
                cn = row.get("Counter_Name") or row.get("Counter Name")
                cv = row.get("Counter_Value") or row.get("Counter Value")
                if not cn:
                    continue
                key = kn.split("(")[0][:60]
                agg[key][cn] += float(cv)
                n[key] += 1
    return agg, n
for tag, pat in [("micro", "gpurun_out/pmc_gemm_micro/**/*counter*.csv"),
                 ("bench", "gpurun_out/pmc_bench/**/*counter*.csv")]:
    agg, n = summarize(pat)
    print(f"==== {tag} ====")
    for k, ctrs in sorted(agg.items()):
        if not any(s in k for s in ("gemm", "Cijk", "attn", "moe")):
            continue
        line = " ".join(f"{c}={v:.3e}" for c, v in sorted(ctrs.items()))
        print(f"{k}: n={n[k]} {line}")
        th = ctrs.get("TCP_UTCL1_TRANSLATION_HIT", 0.0)
        tm = ctrs.get("TCP_UTCL1_TRANSLATION_MISS", 0.0)
        if th + tm > 0:
            print(f"    UTCL1 translation miss-rate: {tm/(th+tm)*100:.3f}%")
        tr = ctrs.get("TCC_REQ", 0.0)
        tc = ctrs.get("TCC_MISS", 0.0)
        if tr > 0:
            print(f"    TCC (L2) miss-rate: {tc/tr*100:.2f}%")
PYEOF
cat gpurun_out/pmc_utcl2_summary.txt
find gpurun_out/pmc_gemm_micro gpurun_out/pmc_bench \
  \( -name '*.csv' -size +8M \) -delete 2>/dev/null
find gpurun_out/pmc_gemm_micro gpurun_out/pmc_bench -name '*.db' -delete 2>/dev/null
du -sh gpurun_out
echo PMC DONE
