#!/bin/bash
# GPU test + bench + profile plan — ONE gpurun call runs this end-to-end:
#   /usr/local/graft/bin/gpurun --timeout 2400 -- 'bash profiles/run_profiles.sh'
# IMPORTANT: gpurun refuses to copy gpurun_out/ back if it exceeds 64 MiB —
# raw rocprof kernel traces are 100s of MB, so every step prunes to the
# small stats/summary files immediately.
set -x
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT="$GRAFT_REPO_ROOT/gpurun_out/prof"
mkdir -p "$OUT"

# -1. MFMA fragment-layout discovery (one-time decode for kernels_mfma.hip)
timeout 300 python profiles/mfma_discover.py > "$OUT"/mfma_layout.txt 2>&1

# 0. correctness first: GPU suite + smoke (no -x: show ALL failures)
timeout 900 python -m pytest tests -m gpu -q > "$OUT"/pytest_gpu.log 2>&1
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
    > "$OUT"/smoke.log 2>&1

# 1. bench (N=1, 256^3) — the BASELINE metric
timeout 1200 python bench.py --gpus 1 --steps 2 --warmup 1 \
    > "$OUT"/bench_256.json 2> "$OUT"/bench_256.err

# 1b. block4 DILU bench config (BASELINE config #4 shape, MFMA path)
timeout 600 python bench.py --gpus 1 --steps 2 --warmup 1 --config block4_dilu \
    > "$OUT"/bench_block4.json 2> "$OUT"/bench_block4.err

# 2. kernel-time table (trace+stats only; NO --pmc here) — keep stats only
cd /tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT"/trace -- \
    python "$GRAFT_REPO_ROOT"/bench.py --gpus 1 --steps 1 --warmup 1 --size 192 \
    > "$OUT"/rocprof_stats.log 2>&1
# prune: keep only *stats* csvs (the raw kernel trace is huge)
find "$OUT"/trace -type f ! -name '*stats*' -delete 2>/dev/null
find "$OUT"/trace -type f -name '*stats*' -size +4M -delete 2>/dev/null

# 2b. kernel microbenchmarks (per-kernel ms + achieved GB/s)
cd "$GRAFT_REPO_ROOT" || cd /root/repo
timeout 600 python bench_kernels.py --size 192 --iters 20 \
    > "$OUT"/kernel_bench.jsonl 2> "$OUT"/kernel_bench.txt

# 3. PMC counters in their OWN run (pool rule: never combined with traces)
cd /tmp
timeout 600 rocprofv3 --pmc FETCH_SIZE SQ_INSTS_MFMA \
    --output-format csv -d "$OUT"/pmc -- \
    python "$GRAFT_REPO_ROOT"/bench.py --gpus 1 --steps 1 --warmup 0 --size 128 \
    > "$OUT"/rocprof_pmc.log 2>&1
# summarize per-kernel counter totals, then drop the raw csv
python - "$OUT"/pmc > "$OUT"/pmc_summary.txt 2>&1 <<'EOF'
import csv, glob, sys, collections
tot = collections.defaultdict(lambda: collections.defaultdict(float))
cnt = collections.Counter()
for f in glob.glob(sys.argv[1] + "/**/*.csv", recursive=True):
    with open(f) as fh:
        for row in csv.DictReader(fh):
            k = row.get("Kernel_Name") or row.get("kernel_name") or "?"
            c = row.get("Counter_Name") or row.get("counter_name")
            v = row.get("Counter_Value") or row.get("counter_value") or 0
            if c:
                tot[k][c] += float(v)
                cnt[k] += 1
for k in sorted(tot, key=lambda k: -tot[k].get("FETCH_SIZE", 0)):
    t = tot[k]
    print(f"{k[:100]} | n={cnt[k]} | " + " ".join(
        f"{c}={t[c]:.3e}" for c in sorted(t)))
EOF
rm -rf "$OUT"/pmc

# 3b. PMC on the block4 DILU config: SQ_INSTS_MFMA must be > 0 now
timeout 600 rocprofv3 --pmc SQ_INSTS_MFMA SQ_INSTS_VALU \
    --output-format csv -d "$OUT"/pmc4 -- \
    python "$GRAFT_REPO_ROOT"/bench.py --gpus 1 --steps 1 --warmup 0 \
    --config block4_dilu --size 128 > "$OUT"/rocprof_pmc4.log 2>&1
python - "$OUT"/pmc4 > "$OUT"/pmc4_summary.txt 2>&1 <<'EOF'
import csv, glob, sys, collections
tot = collections.defaultdict(lambda: collections.defaultdict(float))
for f in glob.glob(sys.argv[1] + "/**/*.csv", recursive=True):
    with open(f) as fh:
        for row in csv.DictReader(fh):
            k = row.get("Kernel_Name") or row.get("kernel_name") or "?"
            c = row.get("Counter_Name") or row.get("counter_name")
            v = row.get("Counter_Value") or row.get("counter_value") or 0
            if c:
                tot[k][c] += float(v)
for k in sorted(tot, key=lambda k: -tot[k].get("SQ_INSTS_MFMA", 0)):
    t = tot[k]
    print(f"{k[:100]} | " + " ".join(f"{c}={t[c]:.3e}" for c in sorted(t)))
EOF
rm -rf "$OUT"/pmc4

# final guard: NOTHING over the pull limit leaves the box
find "$GRAFT_REPO_ROOT"/gpurun_out -type f -size +8M -delete 2>/dev/null
du -sh "$GRAFT_REPO_ROOT"/gpurun_out > "$OUT"/du.txt 2>&1
exit 0
