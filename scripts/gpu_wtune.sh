set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
for cfg in "8:4194304" "8:8388608" "8:16777216" "12:8388608" "16:8388608"; do
  f="${cfg%%:*}"; c="${cfg##*:}"
  timeout 240 python bench.py --workload seqwrite --path client --files $f --read-chunk $c --steps 4 --warmup 1 --hbm-gb 48 > gpurun_out/wt_${f}_${c}.json 2>/dev/null
  echo "f=$f c=$c rc=$?"
done
grep -ho '"value": [0-9.]*' gpurun_out/wt_*.json
exit 0
