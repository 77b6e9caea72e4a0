set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r2.log 2>&1
echo "pytest_gpu rc=$?"
timeout 240 python scripts/meta_bench.py > gpurun_out/meta_native_gpu.json 2>gpurun_out/meta_native_gpu.err
echo "meta_native rc=$?"
timeout 240 python scripts/meta_bench.py --no-native > gpurun_out/meta_asyncio_gpu.json 2>gpurun_out/meta_asyncio_gpu.err
echo "meta_asyncio rc=$?"
timeout 420 python bench.py --steps 30 --warmup 5 > gpurun_out/bench_default_r2.json 2>gpurun_out/bench_default_r2.err
echo "bench_seq rc=$?"
timeout 300 python bench.py --workload randread4k --steps 10 --warmup 2 > gpurun_out/bench_rand4k_r2.json 2>gpurun_out/bench_rand4k_r2.err
echo "bench_rand rc=$?"
timeout 420 python scripts/dataloader_bench.py --workers 8 --to-device > gpurun_out/dl_w8.json 2>gpurun_out/dl_w8.err
echo "dl rc=$?"
tail -3 gpurun_out/pytest_gpu_r2.log
cat gpurun_out/meta_native_gpu.json gpurun_out/meta_asyncio_gpu.json gpurun_out/bench_default_r2.json gpurun_out/bench_rand4k_r2.json gpurun_out/dl_w8.json 2>/dev/null
