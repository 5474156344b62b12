set -x
cd /root/repo
for cfg in "PNR_NOOP=1" "PNR_RESYNC_EVERY=3" "PNR_ASTAR=1.8" "PNR_ASTAR=1.8 PNR_RESYNC_EVERY=3"; do
  echo "===== btc $cfg"
  env $cfg timeout 300 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 2>&1 | grep -oE '"value": [0-9.]+|"wirelength": [0-9]+|"crit_path_ns": [0-9.]+|"route_iterations": [0-9]+' | tr '\n' ' '; echo
done
echo PROBE18 DONE
