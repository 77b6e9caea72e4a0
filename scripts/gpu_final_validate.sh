set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_final.log 2>&1; echo "pytest rc=$?"
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('smoke ok')" > gpurun_out/smoke_final.log 2>&1; echo "smoke rc=$?"
timeout 300 python bench.py --steps 20 --warmup 4 > gpurun_out/bench_final_default.json 2>gpurun_out/bench_final_default.err; echo "bench rc=$?"
timeout 200 python scripts/meta_bench.py > gpurun_out/meta_final.json 2>gpurun_out/meta_final.err; echo "meta rc=$?"
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof_loader" -- python "$GRAFT_REPO_ROOT/scripts/dataloader_bench.py" --device-loader --shards 8 --epochs 1 > "$GRAFT_REPO_ROOT/gpurun_out/prof_loader.log" 2>&1; echo "rocprof rc=$?"
cd "$GRAFT_REPO_ROOT"
tail -2 gpurun_out/pytest_gpu_final.log
tail -2 gpurun_out/smoke_final.log
cat gpurun_out/bench_final_default.json gpurun_out/meta_final.json 2>/dev/null
grep -h "copy_extents\|crc32c\|lz4" gpurun_out/prof_loader/*stats* 2>/dev/null | head -5
