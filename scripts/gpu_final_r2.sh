set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 500 python -m pytest tests -m gpu -q > gpurun_out/final_pytest.log 2>&1; echo "pytest rc=$?"
timeout 240 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/final_smoke.log 2>&1; echo "smoke rc=$?"
timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/final_bench1.json 2>/dev/null; echo "b1 rc=$?"
timeout 180 python bench.py --steps 5 --warmup 2 > gpurun_out/final_bench2.json 2>/dev/null; echo "b2 rc=$?"
timeout 240 python bench.py --workload seqwrite --path client --files 16 --read-chunk 8388608 --steps 4 --warmup 1 --hbm-gb 48 > gpurun_out/final_seqwrite.json 2>/dev/null; echo "w rc=$?"
timeout 240 python bench.py --separate-worker --path client --files 8 --steps 5 --warmup 2 > gpurun_out/final_ipc.json 2>/dev/null; echo "ipc rc=$?"
timeout 240 python bench.py --workload randread4k --path client --steps 5 --warmup 2 > gpurun_out/final_rand4k.json 2>/dev/null; echo "r4k rc=$?"
timeout 240 python scripts/meta_bench.py --n 20000 > gpurun_out/final_meta1.json 2>/dev/null; echo "m1 rc=$?"
timeout 240 python scripts/meta_bench.py --n 20000 > gpurun_out/final_meta2.json 2>/dev/null; echo "m2 rc=$?"
tail -1 gpurun_out/final_pytest.log gpurun_out/final_smoke.log
grep -ho '"value": [0-9.]*' gpurun_out/final_bench1.json gpurun_out/final_bench2.json gpurun_out/final_seqwrite.json gpurun_out/final_ipc.json gpurun_out/final_rand4k.json
cat gpurun_out/final_meta1.json gpurun_out/final_meta2.json
exit 0
