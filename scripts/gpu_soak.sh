set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
for i in 1 2; do
  timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/soak_pytest_$i.log 2>&1; echo "pytest$i rc=$?"
done
for i in 1 2 3; do
  timeout 300 python bench.py --steps 15 --warmup 3 > gpurun_out/soak_bench_$i.json 2>/dev/null; echo "bench$i rc=$?"
done
export TMPDIR=/tmp; cd /tmp
timeout 300 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_gather" -- python "$GRAFT_REPO_ROOT/scripts/dataloader_bench.py" --device-loader --shards 4 --epochs 1 > "$GRAFT_REPO_ROOT/gpurun_out/pmc_gather.log" 2>&1; echo "pmc rc=$?"
cd "$GRAFT_REPO_ROOT"
tail -1 gpurun_out/soak_pytest_1.log; tail -1 gpurun_out/soak_pytest_2.log
grep -ho '"value": [0-9.]*' gpurun_out/soak_bench_*.json
tail -3 gpurun_out/pmc_gather.log
