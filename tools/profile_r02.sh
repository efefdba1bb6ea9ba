#!/bin/bash
# Round-2 profiling battery (run ON the GPU box via gpurun).
# kernel-trace/--stats passes and --pmc passes are SEPARATE rocprofv3 runs
# (combining them is refused by gpurun / crashes nodes).
set -x
cd /tmp && export TMPDIR=/tmp PYTHONUNBUFFERED=1
P=/root/repo/gpurun_out/prof
mkdir -p $P
ok() { echo "== $1 done rc=$?"; }

timeout 240 rocprofv3 --kernel-trace --stats -d $P/stats_verify -- \
  python /root/repo/prof_target.py > $P/stats_verify.log 2>&1; ok stats_verify
timeout 300 rocprofv3 --kernel-trace --stats -d $P/stats_sighash -- \
  python /root/repo/sighash_bench.py > $P/stats_sighash.log 2>&1; ok stats_sighash
timeout 240 rocprofv3 --pmc FETCH_SIZE SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_WAVE_CYCLES SQ_INSTS_VALU -d $P/pmc_fetch_verify -- \
  python /root/repo/prof_target.py > $P/pmc_fetch_verify.log 2>&1; ok pmc_fetch_verify
timeout 240 rocprofv3 --pmc WRITE_SIZE -d $P/pmc_write_verify -- \
  python /root/repo/prof_target.py > $P/pmc_write_verify.log 2>&1; ok pmc_write_verify
timeout 300 rocprofv3 --pmc FETCH_SIZE -d $P/pmc_fetch_sighash -- \
  python /root/repo/sighash_bench.py > $P/pmc_fetch_sighash.log 2>&1; ok pmc_fetch_sighash
timeout 300 rocprofv3 --pmc WRITE_SIZE -d $P/pmc_write_sighash -- \
  python /root/repo/sighash_bench.py > $P/pmc_write_sighash.log 2>&1; ok pmc_write_sighash

python /root/repo/tools/agg_prof.py $P > $P/summary.json 2> $P/agg.err
echo "=== summary head ==="
head -c 3000 $P/summary.json
echo; echo "=== logs tails ==="
tail -3 $P/stats_verify.log $P/stats_sighash.log
