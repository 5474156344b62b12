set -x
cd /root/repo
echo "===== flow-btc"
PNR_ATTEMPT_LOG=1 timeout 400 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=" | tail -42
echo "===== btc rc=$?"
echo "===== flow-sv2"
timeout 240 python bench.py --config stereovision2 --fill 0.5 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\]" | tail -6
echo "===== sv2 rc=$?"
echo "===== flow-bgm"
timeout 240 python bench.py --config bgm --fill 0.5 --steps 1 --warmup 0 --verbose 2>&1 | grep -vE "^\[gpu\] T=|\[dist\]" | tail -6
echo "===== bgm rc=$?"
echo "===== at-scale-quality"
timeout 900 python -m pytest tests/test_gpu_router.py -q -k "at_scale" 2>&1 | tail -4
echo "===== quality rc=$?"
echo PROBE11 DONE
