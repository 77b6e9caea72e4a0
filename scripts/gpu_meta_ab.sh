set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 200 python scripts/meta_bench.py > gpurun_out/ab_native.json 2>gpurun_out/ab_native.err; echo "native rc=$?"
timeout 200 python scripts/meta_bench.py --no-native > gpurun_out/ab_asyncio.json 2>gpurun_out/ab_asyncio.err; echo "asyncio rc=$?"
CURVINE_META_INLINE=0 timeout 200 python scripts/meta_bench.py > gpurun_out/ab_native_queue.json 2>gpurun_out/ab_queue.err; echo "queue rc=$?"
timeout 200 python scripts/meta_bench.py > gpurun_out/ab_native2.json 2>gpurun_out/ab_native2.err; echo "native2 rc=$?"
cat gpurun_out/ab_native.json gpurun_out/ab_asyncio.json gpurun_out/ab_native_queue.json gpurun_out/ab_native2.json 2>/dev/null
