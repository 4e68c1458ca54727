set -x
cd /root/repo
mkdir -p gpurun_out
timeout 800 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r2c.log 2>&1
echo "PYTEST_RC=$?"
tail -3 gpurun_out/pytest_gpu_r2c.log
# e2e with fp8 KV
AURORA_KV_FP8=1 timeout 900 python bench.py --steps 1 --warmup 1 > gpurun_out/bench_e2e_fp8.log 2>&1
echo "FP8_RC=$?"
tail -1 gpurun_out/bench_e2e_fp8.log
