set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_w.log 2>&1
echo "pytest_gpu rc=$?"
timeout 240 python bench.py --workload seqwrite --path client --files 8 --steps 5 --warmup 2 > gpurun_out/bench_seqwrite.json 2>gpurun_out/bench_seqwrite.err
echo "seqwrite rc=$?"
timeout 240 python bench.py --workload seqwrite --path client --files 16 --threads 16 --steps 5 --warmup 2 --hbm-gb 48 > gpurun_out/bench_seqwrite16.json 2>gpurun_out/bench_seqwrite16.err
echo "seqwrite16 rc=$?"
timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/bench_seqread_check.json 2>gpurun_out/bench_seqread_check.err
echo "seqread rc=$?"
timeout 240 python scripts/meta_bench.py --n 20000 > gpurun_out/meta_n20k_a.json 2>gpurun_out/meta_n20k_a.err
echo "meta_a rc=$?"
timeout 240 python scripts/meta_bench.py --n 20000 > gpurun_out/meta_n20k_b.json 2>gpurun_out/meta_n20k_b.err
echo "meta_b rc=$?"
tail -3 gpurun_out/pytest_gpu_w.log
cat gpurun_out/bench_seqwrite.json gpurun_out/bench_seqwrite16.json gpurun_out/bench_seqread_check.json gpurun_out/meta_n20k_a.json gpurun_out/meta_n20k_b.json 2>/dev/null
tail -2 gpurun_out/bench_seqwrite.err gpurun_out/bench_seqwrite16.err 2>/dev/null
