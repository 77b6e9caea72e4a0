set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 300 python bench.py --path client --steps 10 --warmup 3 --read-chunk 4194304 --files 16 --hbm-gb 24 --threads 32 --seq-batch 1 > gpurun_out/peak_4m_b1.json 2>gpurun_out/peak_4m_b1.err; echo "b1 rc=$?"
timeout 300 python bench.py --path client --steps 10 --warmup 3 --read-chunk 4194304 --files 16 --hbm-gb 24 --threads 32 --seq-batch 4 > gpurun_out/peak_4m_b4.json 2>gpurun_out/peak_4m_b4.err; echo "b4 rc=$?"
timeout 300 python bench.py --workload randread4k --steps 8 --warmup 2 --iodepth 128 --threads 32 > gpurun_out/rand_d128_t32.json 2>gpurun_out/rand_d128_t32.err; echo "rand rc=$?"
timeout 420 python scripts/dataloader_bench.py --device-loader --batch-size 512 > gpurun_out/dl_b512.json 2>gpurun_out/dl_b512.err; echo "dl rc=$?"
cat gpurun_out/peak_4m_b1.json gpurun_out/peak_4m_b4.json gpurun_out/rand_d128_t32.json gpurun_out/dl_b512.json 2>/dev/null
