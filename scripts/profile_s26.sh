#!/bin/bash
# Profiling recipes for the GPU box (run via gpurun). Round-2 state.
# Kernel-time stats (trace run — NEVER combine --pmc with trace domains;
# write rocprof output to /tmp, copy only the small stats CSV back):
#   export TMPDIR=/tmp && cd /tmp
#   rocprofv3 --output-format csv --kernel-trace --stats -d /tmp/prof -o run -- \
#       python /root/repo/bench.py --scale 26 --steps 3 --warmup 1 --no-converge
#   cp /tmp/prof/run_kernel_stats.csv /root/repo/gpurun_out/
# PMC counters (separate run, no trace flags; SQ=8/TCC=4 slots per pass):
#   rocprofv3 --output-format csv --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY \
#       SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_INSTS_VALU \
#       SQ_LDS_BANK_CONFLICT FETCH_SIZE -d /tmp/pmc -o run -- \
#       python /root/repo/bench.py --scale 22 --steps 2 --warmup 1 --no-converge
# Per-class sync timing + knob A/Bs:
#   CUVITE_PROGRESS=1 python bench.py --scale 26 --steps 2 --warmup 1 --no-converge
#   CUVITE_HUB_CUT=4096  python bench.py --scale 26 --steps 10 --warmup 3  # ref cut
#   CUVITE_HUB_SEGSORT=0 python bench.py --scale 26 --steps 10 --warmup 3  # torch-sort hub
#   CUVITE_NO_OVERLAP=1  python bench.py --scale 26 --steps 10 --warmup 3  # one stream
