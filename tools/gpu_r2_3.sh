set -x
cd /root/repo
mkdir -p gpurun_out
# e2e bench at the new defaults (B=128, kv 0.8, aux-stream rail/RAG)
timeout 1300 python bench.py --steps 1 --warmup 1 > gpurun_out/bench_e2e_128.log 2>&1
echo "E2E_RC=$?"
grep -E "\[bench\]" gpurun_out/bench_e2e_128.log | tail -6
tail -1 gpurun_out/bench_e2e_128.log
