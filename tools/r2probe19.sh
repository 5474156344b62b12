set -x
cd /root/repo
echo "===== btc astar2.2"
PNR_ASTAR=2.2 timeout 300 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 2>&1 | grep -oE '"value": [0-9.]+|"wirelength": [0-9]+|"crit_path_ns": [0-9.]+' | tr '\n' ' '; echo
for c in LU32PEEng stereovision2 bgm; do
  echo "===== $c default"
  timeout 240 python bench.py --config $c --fill 0.55 --steps 1 --warmup 0 2>&1 | grep -oE '"value": [0-9.]+|"wirelength": [0-9]+|"crit_path_ns": [0-9.]+|"feasible": [a-z]+' | tr '\n' ' '; echo
done
echo "===== at-scale quality @1.8 default pathfinder"
timeout 600 python -m pytest tests/test_gpu_router.py tests/test_gpu_place.py -q 2>&1 | tail -2
echo PROBE19 DONE
