set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_mid.log 2>&1
echo "pytest_gpu rc=$?"
timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/bench_mid.json 2>gpurun_out/bench_mid.err
echo "bench rc=$?"
timeout 300 python scripts/meta_bench.py --n 40000 > gpurun_out/meta_mid.json 2>gpurun_out/meta_mid.err
echo "meta rc=$?"
timeout 300 python scripts/meta_bench.py --n 40000 > gpurun_out/meta_mid2.json 2>gpurun_out/meta_mid2.err
echo "meta2 rc=$?"
tail -3 gpurun_out/pytest_gpu_mid.log
cat gpurun_out/bench_mid.json gpurun_out/meta_mid.json gpurun_out/meta_mid2.json 2>/dev/null
