#!/bin/bash
# Kernel-sanity runs for the HIP kernels (SURVEY.md §5.2 — the reference has
# no race/sanitizer story; this is the flowhip equivalent for ROCm):
#  - AMD_SERIALIZE_KERNEL=3: serialize launches so faults attribute to the
#    offending kernel.
#  - HSA_XNACK=1: page-fault capture instead of silent corruption.
# Run via gpurun: bash tools/sanitize.sh
set -e
REPO=${GRAFT_REPO_ROOT:-/root/repo}
cd "$REPO"

echo "== serialized-kernel GPU test pass =="
AMD_SERIALIZE_KERNEL=3 python -m pytest tests -m gpu -q -x

echo "== xnack smoke =="
HSA_XNACK=1 timeout 300 python bench.py --steps 2 --warmup 1 --profile-steps 1
echo OK
