#!/bin/bash
# Round-2 PMC capture recipe (run via gpurun on an MI355X box):
#   /usr/local/graft/bin/gpurun --timeout 500 -- 'bash tools/profile_pmc.sh'
# Writes per-kernel counter CSVs into gpurun_out/pmc/ (copy the ones to
# keep into profiles/ and commit). Counters and kernel-trace MUST be
# separate runs (gpurun refuses combined --pmc + trace domains).
set -e
export TMPDIR=/tmp
mkdir -p gpurun_out/pmc
timeout 200 rocprofv3 -i tools/pmc_counters.txt --output-format csv \
    -d /tmp/pmc -- env SAGECAL_NO_GRAPH=1 \
    python tools/profile_step.py --tilesz 60 --reps 1 > /tmp/pmc.log 2>&1
find /tmp/pmc -name "*.csv" -exec cp {} gpurun_out/pmc/ \;
tail -3 /tmp/pmc.log
ls gpurun_out/pmc/
