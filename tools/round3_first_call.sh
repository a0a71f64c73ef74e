#!/bin/bash
# Round-3 first GPU call (run via gpurun, ~12 min):
#   /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/round3_first_call.sh'
# 1. validate the two gated round-2 kernels (then flip defaults on)
# 2. pinpoint the config-5 generation wall (per-op log survives timeout)
# 3. fresh headline bench + rocprof kernel stats as FILES into gpurun_out
set -x
mkdir -p gpurun_out
# 1) gated kernels + full GPU suite
SAGECAL_CHOL_BIG=1 SAGECAL_BEAM_KERNEL_TEST=1 \
    timeout 300 python -m pytest tests -m gpu -q 2>&1 | tail -4
# 2) config-5 generation wall, op by op (log flushes incrementally)
rm -f gpurun_out/ska_stages.log
timeout 240 python tools/ska_stages.py --phase gen --stations 512 \
    --dirs 20 2>&1 | tail -2
# also A/B the kernel-backed generation
SAGECAL_GEN_KERNEL=1 timeout 120 python tools/ska_stages.py --phase gen \
    --stations 512 --dirs 20 2>&1 | tail -2
cat gpurun_out/ska_stages.log
# 3) headline bench + rocprof stats (files, not stdout)
timeout 180 python bench.py --steps 3 --warmup 1 2>/dev/null \
    | tee gpurun_out/bench_r3_headline.json | tail -1
cd /tmp && export TMPDIR=/tmp
timeout 200 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /root/repo/gpurun_out/prof -o r3 -- \
    python /root/repo/bench.py --steps 1 --warmup 1 >/tmp/p.log 2>&1
tail -2 /tmp/p.log
ls /root/repo/gpurun_out/prof 2>/dev/null | head
# 4) config-5 bench (generation now on GPU after the threshold fix)
cd /root/repo
timeout 300 python bench.py --stations 512 --dirs 20 --solver rtr \
    --shapelet-dirs 2 --intervals 1 --emiter 2 --steps 2 --warmup 1 \
    2>/dev/null | tee gpurun_out/bench_config5.json | tail -1
