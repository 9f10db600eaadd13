#!/bin/bash
# rocprofv3 wrapper for bench runs (counterpart of the reference's nsys
# wrapper benchmarks/bench.sh): profiles rank 0 (and the last rank when
# launched under torchrun) with --kernel-trace --stats.
# Usage: bench_profile.sh <outdir> <python args...>
set -e
outdir=${1:?usage: bench_profile.sh <outdir> <python args...>}
shift
rank=${RANK:-0}
world=${WORLD_SIZE:-1}
last=$((world - 1))
mkdir -p "$outdir"
export TMPDIR=${TMPDIR:-/tmp}
if [ "$rank" = "0" ] || [ "$rank" = "$last" ]; then
  exec rocprofv3 --kernel-trace --stats -d "$outdir" -o "rank${rank}" \
      --output-format csv -- python "$@"
else
  exec python "$@"
fi
