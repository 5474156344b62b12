set -x
cd /root/repo
python -m pytest tests/ -q -m gpu -k "not at_scale" -x 2>&1 | tail -3
echo "===== flow-lu32"
timeout 300 python bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -20
echo "===== lu32 rc=$?"
echo "===== flow-btc"
timeout 700 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -20
echo "===== btc rc=$?"
echo PROBE5 DONE
