set -x
cd /root/repo
python -m pytest tests/ -q -m gpu -k "not at_scale" -x 2>&1 | tail -4
echo "===== flow-lu32"
timeout 480 python bench.py --config LU32PEEng --fill 0.55 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -40
echo "===== flow-lu32 rc=$?"
echo "===== flow-btc"
timeout 900 python bench.py --config bitcoin_miner --fill 0.6 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -40
echo "===== flow-btc rc=$?"
echo "===== ws2-gloo-1gpu"
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29533 bench.py --gpus 2 --config stereovision2 --fill 0.5 --steps 1 --warmup 0 --verbose 2>&1 | grep -v "^\[gpu\] T=" | tail -25
echo "===== ws2 rc=$?"
echo PROBE3 DONE
