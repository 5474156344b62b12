set -x
cd /root/repo
echo "===== full gpu suite"
timeout 700 python -m pytest tests/ -q -m gpu 2>&1 | tail -3
echo "===== suite rc=$?"
echo "===== smoke"
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -2
echo "===== smoke rc=$?"
echo "===== bench default (driver invocation)"
timeout 500 python bench.py 2>&1 | tail -2
echo "===== bench rc=$?"
echo FINAL DONE
