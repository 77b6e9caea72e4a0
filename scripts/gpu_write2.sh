set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_w2.log 2>&1
echo "pytest_gpu rc=$?"
timeout 240 python bench.py --workload seqwrite --path client --files 8 --steps 5 --warmup 2 > gpurun_out/w2_seqwrite8.json 2>gpurun_out/w2_seqwrite8.err
echo "sw8 rc=$?"
timeout 240 python bench.py --workload seqwrite --path client --files 16 --steps 5 --warmup 2 --hbm-gb 48 > gpurun_out/w2_seqwrite16.json 2>gpurun_out/w2_seqwrite16.err
echo "sw16 rc=$?"
timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/w2_seqread.json 2>gpurun_out/w2_seqread.err
echo "sr rc=$?"
timeout 240 python bench.py --workload randread4k --path client --steps 5 --warmup 2 > gpurun_out/w2_rand4k.json 2>gpurun_out/w2_rand4k.err
echo "r4k rc=$?"
tail -3 gpurun_out/pytest_gpu_w2.log
cat gpurun_out/w2_*.json 2>/dev/null
exit 0
