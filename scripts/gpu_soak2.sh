set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/soak_pytest1.log 2>&1; echo "p1 rc=$?"
for i in 1 2 3; do
  timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/soakA_$i.json 2>/dev/null; echo "rA$i=$?"
  timeout 240 python bench.py --workload seqwrite --path client --files 8 --steps 4 --warmup 1 > gpurun_out/soakW_$i.json 2>/dev/null; echo "rW$i=$?"
done
timeout 300 python bench.py --workload randread4k --path client --steps 5 --warmup 2 > gpurun_out/soakR4k.json 2>/dev/null; echo "r4k=$?"
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/soak_pytest2.log 2>&1; echo "p2 rc=$?"
timeout 300 python scripts/meta_bench.py --n 40000 > gpurun_out/soak_meta.json 2>/dev/null; echo "meta=$?"
grep -ho '"value": [0-9.]*' gpurun_out/soakA_*.json gpurun_out/soakW_*.json gpurun_out/soakR4k.json
tail -1 gpurun_out/soak_pytest1.log gpurun_out/soak_pytest2.log
cat gpurun_out/soak_meta.json
exit 0
