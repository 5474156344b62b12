set -x
cd /root/repo
mkdir -p gpurun_out
echo "===== btc-with-128-large"
timeout 400 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\]|attempt" | tail -6
echo "===== btc rc=$?"
export TMPDIR=/tmp
cd /tmp
echo "===== rocprof-lu32"
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof -o lu32 -- python /root/repo/bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 2>&1 | tail -3
echo "===== rocprof rc=$?"
find /tmp/prof -name "*stats*" -o -name "*.csv" | head -10
for f in $(find /tmp/prof -name "*kernel_stats*"); do cp "$f" /root/repo/gpurun_out/; done
ls -la /root/repo/gpurun_out/*.csv 2>/dev/null
echo PROBE13 DONE
