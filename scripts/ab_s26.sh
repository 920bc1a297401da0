#!/bin/bash
# Round-2 A/B matrix at R-MAT s26, 1 GPU. Run on the GPU box:
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash scripts/ab_s26.sh'
# Writes one log per arm under gpurun_out/ab/ and prints a summary line each.
set -x
mkdir -p gpurun_out/ab

run() {
  name=$1
  envs=$2
  extra=$3
  timeout 300 env $envs python bench.py --scale 26 --steps 10 --warmup 3 \
      $extra > "gpurun_out/ab/$name.log" 2>&1
  echo "$name exit=$? $(grep -o '"ms_per_step": [0-9.]*' \
      "gpurun_out/ab/$name.log" | head -1)"
}

run baseline     "" ""
run segsort      "CUVITE_HUB_SEGSORT=1" ""
run no_overlap   "CUVITE_NO_OVERLAP=1" ""
run segsort_noov "CUVITE_HUB_SEGSORT=1 CUVITE_NO_OVERLAP=1" ""
run fp32         "" "--dtype fp32"
run alltoall     "CUVITE_TRANSPORT=alltoall" ""   # same as baseline at N=1

grep -H '"ms_per_step"' gpurun_out/ab/*.log | sed 's/.*ab\///'
