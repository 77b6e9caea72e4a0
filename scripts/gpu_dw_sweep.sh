set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
for cfg in "4:4194304" "8:4194304" "8:8388608" "16:8388608"; do
  w="${cfg%%:*}"; c="${cfg##*:}"
  CURVINE_DW_WINDOW=$w CURVINE_DW_CHUNK=$c timeout 240 python bench.py --workload seqwrite --path client --files 8 --steps 4 --warmup 1 --no-short-circuit > gpurun_out/dw_${w}_${c}.json 2>/dev/null
  echo "w=$w c=$c rc=$?"
done
grep -h '"value"' gpurun_out/dw_*.json | grep -o '"value": [0-9.]*'
exit 0
