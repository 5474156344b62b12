set -x
cd /root/repo
python -m pytest tests/ -q -m gpu -k "not at_scale" -x 2>&1 | tail -3
echo "===== flow-lu32"
PNR_ATTEMPT_LOG=1 timeout 240 python bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=" | tail -20
echo "===== lu32 rc=$?"
echo "===== flow-btc"
PNR_ATTEMPT_LOG=1 timeout 500 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=" | tail -45
echo "===== btc rc=$?"
echo PROBE10 DONE
