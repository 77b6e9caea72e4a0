set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r3.log 2>&1
echo "pytest_gpu rc=$?"
timeout 240 python scripts/meta_bench.py > gpurun_out/meta_native_gpu2.json 2>gpurun_out/meta_native_gpu2.err
echo "meta_native rc=$?"
timeout 420 python scripts/dataloader_bench.py --device-loader > gpurun_out/dl_device.json 2>gpurun_out/dl_device.err
echo "dl_device rc=$?"
timeout 420 python scripts/dataloader_bench.py --device-loader --batch-size 256 > gpurun_out/dl_device_b256.json 2>gpurun_out/dl_device_b256.err
echo "dl_device_b256 rc=$?"
timeout 300 python bench.py --workload randread4k --steps 10 --warmup 2 > gpurun_out/bench_rand4k_r3.json 2>gpurun_out/bench_rand4k_r3.err
echo "bench_rand rc=$?"
tail -3 gpurun_out/pytest_gpu_r3.log
cat gpurun_out/meta_native_gpu2.json gpurun_out/dl_device.json gpurun_out/dl_device_b256.json gpurun_out/bench_rand4k_r3.json 2>/dev/null
