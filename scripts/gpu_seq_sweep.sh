set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r4.log 2>&1; echo "pytest rc=$?"
for B in 1 4 8; do
  timeout 300 python bench.py --path client --steps 12 --warmup 3 --seq-batch $B > gpurun_out/seqb_$B.json 2>gpurun_out/seqb_$B.err
  echo "seqb$B rc=$?"
done
timeout 300 python bench.py --path client --steps 12 --warmup 3 --seq-batch 4 --read-chunk 4194304 --files 16 > gpurun_out/seqb4_4m.json 2>gpurun_out/seqb4_4m.err
echo "seqb4_4m rc=$?"
tail -2 gpurun_out/pytest_gpu_r4.log
cat gpurun_out/seqb_1.json gpurun_out/seqb_4.json gpurun_out/seqb_8.json gpurun_out/seqb4_4m.json 2>/dev/null
