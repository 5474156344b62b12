set -x
cd /root/repo
echo "===== btc-noresync"
PNR_RESYNC_EVERY=0 PNR_ATTEMPT_LOG=1 timeout 600 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -45
echo "===== btc-noresync rc=$?"
echo "===== rocprof-lu32"
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_lu32 -o lu32 -- python /root/repo/bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 2>&1 | tail -8
echo "===== rocprof rc=$?"
ls /root/repo/gpurun_out/prof_lu32/ 2>/dev/null
echo PROBE6 DONE
