#!/bin/bash
# Round-2 GPU profiling plan — ONE gpurun call runs this end-to-end:
#   /usr/local/graft/bin/gpurun --timeout 2400 -- 'bash profiles/run_profiles.sh'
# Writes everything under gpurun_out/prof; copy the summaries worth keeping
# into profiles/ and commit them.
set -x
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out/prof

# 0. correctness first: GPU suite + smoke (no -x: show ALL failures)
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/prof/pytest_gpu.log 2>&1
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/prof/smoke.log 2>&1

# 1. bench (N=1, 256^3) — the BASELINE metric
timeout 1200 python bench.py --gpus 1 --steps 3 --warmup 1 \
    > gpurun_out/prof/bench_256.json 2> gpurun_out/prof/bench_256.err

# 2. kernel-time table for the same run (trace+stats only; NO --pmc here)
cd /tmp && timeout 900 rocprofv3 --kernel-trace --stats \
    -d "$GRAFT_REPO_ROOT"/gpurun_out/prof/trace -- \
    python "$GRAFT_REPO_ROOT"/bench.py --gpus 1 --steps 1 --warmup 1 --size 192 \
    > "$GRAFT_REPO_ROOT"/gpurun_out/prof/rocprof_stats.log 2>&1

# 2b. kernel microbenchmarks (per-kernel ms + achieved GB/s)
cd "$GRAFT_REPO_ROOT" || cd /root/repo
timeout 600 python bench_kernels.py --size 192 --iters 20 \
    > gpurun_out/prof/kernel_bench.jsonl 2> gpurun_out/prof/kernel_bench.txt

# 3. PMC counters in their OWN run (pool rule: never combined with traces)
cd /tmp
timeout 900 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE SQ_INSTS_MFMA \
    -d "$GRAFT_REPO_ROOT"/gpurun_out/prof/pmc -- \
    python "$GRAFT_REPO_ROOT"/bench.py --gpus 1 --steps 1 --warmup 0 --size 128 \
    > "$GRAFT_REPO_ROOT"/gpurun_out/prof/rocprof_pmc.log 2>&1
exit 0
