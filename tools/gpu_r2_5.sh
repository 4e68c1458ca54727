set -x
cd /root/repo
mkdir -p gpurun_out
timeout 700 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r2d.log 2>&1
echo "PYTEST_RC=$?"
tail -2 gpurun_out/pytest_gpu_r2d.log
# engine-mode regressions vs the r01 headline (same config: B=256, 4 turns)
timeout 300 python bench.py --mode engine --concurrency 256 --steps 2 --warmup 1 > gpurun_out/bench_engine_8b.log 2>&1
echo "ENG8B_RC=$?"; tail -1 gpurun_out/bench_engine_8b.log
BENCH_MODEL=mixtral-8x7b timeout 420 python bench.py --mode engine --concurrency 256 --steps 2 --warmup 1 > gpurun_out/bench_engine_mixtral.log 2>&1
echo "MIX_RC=$?"; tail -1 gpurun_out/bench_engine_mixtral.log
# fp8 e2e retest with the exact-truncation conversion
AURORA_KV_FP8=1 timeout 800 python bench.py --steps 1 --warmup 1 > gpurun_out/bench_e2e_fp8b.log 2>&1
echo "FP8B_RC=$?"; tail -1 gpurun_out/bench_e2e_fp8b.log
