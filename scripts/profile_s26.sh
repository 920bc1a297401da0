#!/bin/bash
# Round-2 profiling recipes for the GPU box (run via gpurun).
# Kernel-time stats (trace run -- NEVER combine --pmc with trace domains):
#   cd /tmp && export TMPDIR=/tmp
#   rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/profNN -- \
#       python /root/repo/bench.py --scale 26 --steps 5 --warmup 2
# PMC counters (separate run, no trace flags):
#   rocprofv3 --pmc SQ_WAVES,SQ_INSTS_LDS,SQ_LDS_BANK_CONFLICT \
#       -d /root/repo/gpurun_out/pmcNN -- \
#       python /root/repo/bench.py --scale 26 --steps 3 --warmup 1
#   rocprofv3 --pmc FETCH_SIZE,WRITE_SIZE -d ... -- <same>
# Hub-path A/B:
#   CUVITE_PROGRESS=1 python bench.py --scale 26 --steps 2 --warmup 1   # per-class timing
#   CUVITE_HUB_SEGSORT=1 python bench.py --scale 26 --steps 10 --warmup 3
#   CUVITE_NO_OVERLAP=1 python bench.py --scale 26 --steps 10 --warmup 3
# Summarize a trace db:
#   python /root/repo/profiles/summarize.py gpurun_out/profNN/runc/*_results.db
