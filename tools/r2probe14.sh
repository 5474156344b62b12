set -x
cd /root/repo
echo "===== dist-gpu-test"
timeout 600 python -m pytest tests/test_dist_gpu.py -q 2>&1 | tail -4
echo "===== dist rc=$?"
export TMPDIR=/tmp
cd /tmp
echo "===== rocprof-lu32-stats"
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof -o lu32 -- python /root/repo/bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 > /tmp/prof_stdout.log 2>&1
echo rc=$?
grep -A 25 "KERNEL_DISPATCH\|NAME.*CALLS\|Kernel Name" /tmp/prof_stdout.log | head -40
tail -5 /tmp/prof_stdout.log
cp /tmp/prof_stdout.log /root/repo/gpurun_out/rocprof_lu32_stdout.log
find /tmp/prof -type f | head; du -sh /tmp/prof
echo PROBE14 DONE
