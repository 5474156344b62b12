set -x
cd /root/repo
echo "===== flow-btc-timed"
timeout 700 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -30
echo "===== rc=$?"
echo PROBE4 DONE
