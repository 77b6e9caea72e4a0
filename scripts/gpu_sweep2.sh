set -x
cd "$GRAFT_REPO_ROOT"
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 300 python bench.py --workload randread4k --steps 8 --warmup 2 --threads 32 --iodepth 128 > gpurun_out/rand_t32_d128_reg.json 2>/dev/null; echo r1=$?
timeout 300 python bench.py --workload randread4k --steps 8 --warmup 2 --threads 32 --iodepth 64 > gpurun_out/rand_t32_d64_reg.json 2>/dev/null; echo r2=$?
timeout 300 python bench.py --path client --steps 10 --warmup 3 --read-chunk 4194304 --files 16 --hbm-gb 24 --threads 32 --seq-batch 2 > gpurun_out/seq_4m_b2_reg.json 2>/dev/null; echo s1=$?
timeout 300 python bench.py --path client --steps 10 --warmup 3 --read-chunk 4194304 --files 16 --hbm-gb 24 --threads 32 --seq-batch 4 > gpurun_out/seq_4m_b4_reg.json 2>/dev/null; echo s2=$?
cat gpurun_out/rand_t32_d128_reg.json gpurun_out/rand_t32_d64_reg.json gpurun_out/seq_4m_b2_reg.json gpurun_out/seq_4m_b4_reg.json 2>/dev/null
