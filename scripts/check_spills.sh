#!/usr/bin/env bash
# Scratch-spill audit for every HIP kernel (no GPU needed — hipcc
# cross-compiles). A nonzero ScratchSize means the compiler is spilling
# registers to scratch memory EVERY iteration of whatever loop holds
# the spilled values; ln_bwd_k lost 2.1x to exactly this (DESIGN.md §5:
# missing __launch_bounds__ capped VGPRs at 128). Run after adding or
# restructuring any kernel; the build must stay spill-free.
#
#   bash scripts/check_spills.sh           # audit, exit 1 on any spill
set -u
cd "$(dirname "$0")/../pytorch_distributed_train_amd/ops/csrc"
rc=0
for f in *.hip; do
  out=$(/opt/rocm/bin/hipcc --offload-arch=gfx950 -O3 -c "$f" \
        -o /tmp/spillcheck.o -Rpass-analysis=kernel-resource-usage 2>&1 |
        awk '/Function Name/{fn=$5} /ScratchSize/{if ($3+0 > 0) print fn, $3}')
  if [ -n "$out" ]; then
    echo "SPILLS in $f:"
    echo "$out"
    rc=1
  fi
done
[ $rc -eq 0 ] && echo "no scratch spills in any kernel"
exit $rc
