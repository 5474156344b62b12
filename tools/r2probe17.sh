set -x
cd /root/repo
for cfg in "PNR_RESYNC_EVERY=0" "PNR_RESYNC_EVERY=3" "PNR_ASTAR=1.5" "PNR_DELTA=6.0" "PNR_ASTAR=1.5 PNR_DELTA=6.0"; do
  echo "===== btc $cfg"
  env $cfg timeout 300 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -E '"value"' | tail -1
done
echo PROBE17 DONE
