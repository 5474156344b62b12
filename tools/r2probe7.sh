set -x
cd /root/repo
echo "===== flow-lu32"
PNR_ATTEMPT_LOG=1 timeout 240 python bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\] iter" | tail -12
echo "===== lu32 rc=$?"
echo "===== btc-resync2"
PNR_ATTEMPT_LOG=1 timeout 500 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=" | tail -40
echo "===== btc2 rc=$?"
echo "===== btc-resync3"
PNR_RESYNC_EVERY=3 timeout 500 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\]" | tail -8
echo "===== btc3 rc=$?"
echo PROBE7 DONE
