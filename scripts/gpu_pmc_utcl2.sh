#!/bin/bash
# UTCL2/TLB + L2 PMC probe for the in-context decode-GEMM parking question
# (ROUND3.md item 1): same counters on (a) the standalone GEMM micro and
# (b) the full bench.py decode step, compare gemm_skinny miss rates.
set -x
mkdir -p gpurun_out
cd /root/repo

rocprofv3 -L > gpurun_out/pmc_counters_list.txt 2>&1
grep -oE '[A-Z][A-Z0-9_]+' gpurun_out/pmc_counters_list.txt | sort -u \
  | grep -E 'UTCL|TCC_(REQ|MISS|HIT|EA_RDREQ|EA)' | head -40 \
  > gpurun_out/pmc_candidates.txt
cat gpurun_out/pmc_candidates.txt

pick() {  # first candidate that matches regex $1
  grep -m1 -E "$1" gpurun_out/pmc_candidates.txt
}
C_UTC_REQ=$(pick '^UTCL2.*(REQ|REQUEST)')
C_UTC_MISS=$(pick '^UTCL2.*MISS')
C_TCC_REQ=$(pick '^TCC_REQ')
C_TCC_MISS=$(pick '^TCC_MISS')
CTRS=$(echo "$C_UTC_REQ $C_UTC_MISS $C_TCC_REQ $C_TCC_MISS" | xargs)
echo "chosen counters: $CTRS"
[ -z "$CTRS" ] && { echo "no counters found"; exit 0; }

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
    for f in glob.glob(pat):
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
        for rq, ms in (("UTCL2", "UTCL2"), ("TCC", "TCC")):
            reqs = [v for c, v in ctrs.items() if c.startswith(rq) and ("REQ" in c)]
            miss = [v for c, v in ctrs.items() if c.startswith(ms) and "MISS" in c]
            if reqs and miss and reqs[0] > 0:
                print(f"    {rq} miss-rate: {miss[0]/reqs[0]*100:.2f}%")
PYEOF
cat gpurun_out/pmc_utcl2_summary.txt
find gpurun_out/pmc_gemm_micro gpurun_out/pmc_bench -name '*.csv' -size +8M -delete
find gpurun_out/pmc_gemm_micro gpurun_out/pmc_bench -name '*.db' -delete 2>/dev/null
du -sh gpurun_out
echo PMC DONE
