set -x
cd /root/repo
python -m pytest tests/ -q -m gpu -x 2>&1 | tail -4
echo "===== gpu-suite rc=$?"
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof2 -o lu32 -- python /root/repo/bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 > /tmp/ps.log 2>&1
echo "rocprof rc=$?"
find /tmp/prof2 -type f
for f in $(find /tmp/prof2 -name "*stats*"); do cp "$f" /root/repo/gpurun_out/; done
head -15 /tmp/prof2/*kernel_stats* 2>/dev/null
echo PROBE15 DONE
