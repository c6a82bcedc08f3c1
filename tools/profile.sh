#!/bin/bash
# rocprofv3 capture recipes for the MI355X box (run via gpurun).
#
# IMPORTANT pool rule: never combine --pmc (or -i counter files) with
# -s/--sys-trace, -r/--runtime-trace or hip/hsa/memory-copy/scratch-memory/
# marker trace domains in ONE invocation — collect counters in their own run.
#
# Usage:  bash tools/profile.sh trace   [out_dir]   # kernel trace + stats
#         bash tools/profile.sh pmc     [out_dir]   # MFMA/LDS/HBM counters
#         bash tools/profile.sh infer   [out_dir]   # inference trace
set -e
MODE=${1:-trace}
OUT=${2:-gpurun_out/prof_$MODE}
cd /tmp && export TMPDIR=/tmp
REPO=${GRAFT_REPO_ROOT:-/root/repo}
mkdir -p "$REPO/$OUT"

case "$MODE" in
  trace)
    # FLOWHIP_MIOPEN_FIND=0: the trace covers the whole process including
    # warmup; benchmark-mode MIOpen find would dominate it with solver
    # trials (throughput is identical either way — bench10 A/B)
    FLOWHIP_MIOPEN_FIND=0 \
    rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o step \
      -- python "$REPO/bench.py" --steps 4 --warmup 8 --profile-steps 4
    cp /tmp/prof/*stats*.csv "$REPO/$OUT/" 2>/dev/null || true
    ;;
  pmc)
    # counters only (no trace domains!); SQ has 8 slots, TCC 4.
    # The per-dispatch CSV is ~100 MB — summarize on the box and ship only
    # the per-kernel aggregate (gpurun merge cap is 64 MiB).
    FLOWHIP_MIOPEN_FIND=0 \
    rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
      SQ_LDS_BANK_CONFLICT SQ_VALU_MFMA_BUSY_CYCLES \
      --output-format csv -d /tmp/pmc -o step \
      -- python "$REPO/bench.py" --steps 2 --warmup 6 --profile-steps 2
    python "$REPO/tools/pmc_summarize.py" /tmp/pmc/*counter_collection*.csv \
      > "$REPO/$OUT/pmc_summary.csv"
    cp /tmp/pmc/*agent_info*.csv "$REPO/$OUT/" 2>/dev/null || true
    ;;
  infer)
    rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof -o infer \
      -- python "$REPO/tools/bench_infer.py"
    cp /tmp/prof/*stats*.csv "$REPO/$OUT/" 2>/dev/null || true
    ;;
  *)
    echo "unknown mode $MODE"; exit 1;;
esac
ls -la "$REPO/$OUT/"
