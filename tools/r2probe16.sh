set -x
cd /root/repo
python -m pytest tests/test_gpu_place.py tests/test_gpu_router.py -q -x -k "not at_scale" 2>&1 | tail -3
echo "===== place-tests rc=$?"
echo "===== flow-lu32"
timeout 240 python bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\]" | tail -5
echo "===== lu32 rc=$?"
echo "===== flow-btc"
timeout 400 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\]" | tail -5
echo "===== btc rc=$?"
echo PROBE16 DONE
