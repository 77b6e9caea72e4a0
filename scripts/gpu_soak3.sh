set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
for i in 1 2 3 4 5; do
  timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/s3_read_$i.json 2>/dev/null; echo "r$i=$?"
done
for i in 1 2; do
  timeout 240 python bench.py --workload seqwrite --path client --files 16 --read-chunk 8388608 --steps 4 --warmup 1 --hbm-gb 48 > gpurun_out/s3_write_$i.json 2>/dev/null; echo "w$i=$?"
  timeout 240 python bench.py --separate-worker --path client --files 8 --steps 5 --warmup 2 > gpurun_out/s3_ipc_$i.json 2>/dev/null; echo "i$i=$?"
done
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/s3_pytest.log 2>&1; echo "p=$?"
grep -ho '"value": [0-9.]*' gpurun_out/s3_*.json
tail -1 gpurun_out/s3_pytest.log
exit 0
