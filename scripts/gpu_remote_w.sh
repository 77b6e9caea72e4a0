set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_rw.log 2>&1
echo "pytest rc=$?"
timeout 300 python bench.py --workload seqwrite --path client --files 8 --steps 5 --warmup 2 --no-short-circuit > gpurun_out/rw_write_nosc.json 2>gpurun_out/rw_write_nosc.err
echo "w_nosc rc=$?"
timeout 300 python bench.py --files 8 --steps 5 --warmup 2 --no-short-circuit --read-chunk 4194304 > gpurun_out/rw_read_nosc.json 2>gpurun_out/rw_read_nosc.err
echo "r_nosc rc=$?"
tail -3 gpurun_out/pytest_gpu_rw.log
cat gpurun_out/rw_write_nosc.json gpurun_out/rw_read_nosc.json 2>/dev/null
tail -3 gpurun_out/rw_write_nosc.err 2>/dev/null
exit 0
