set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
export TMPDIR=/tmp
mkdir -p gpurun_out/prof_w
cd /tmp && cd "$GRAFT_REPO_ROOT"
timeout 300 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_w -o seqwrite -- python bench.py --workload seqwrite --path client --files 8 --steps 3 --warmup 1 > gpurun_out/prof_seqwrite.log 2>&1
echo "rocprof rc=$?"
for i in 1 2 3; do
  timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/soak2_$i.json 2>/dev/null
  echo "soak$i rc=$?"
done
grep -h metric gpurun_out/soak2_*.json
tail -4 gpurun_out/prof_seqwrite.log
ls gpurun_out/prof_w | head
exit 0
