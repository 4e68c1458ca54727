set -x
cd /root/repo
mkdir -p gpurun_out
# 1. GPU test suite (includes the one-shot allreduce protocol test)
timeout 1000 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r2b.log 2>&1
echo "PYTEST_RC=$?"
tail -4 gpurun_out/pytest_gpu_r2b.log
# 2. skinny v3 kernel: numerics + cold-weight sweep vs blaslt/v2m
timeout 600 python tools/sweep_skinny3.py > gpurun_out/sweep_skinny3.log 2>&1
echo "SWEEP_RC=$?"
cat gpurun_out/sweep_skinny3.log
# 3. e2e bench at B=256 turns=4: one timed step, phase counters
timeout 1200 python bench.py --steps 1 --warmup 1 > gpurun_out/bench_e2e_256.log 2>&1
echo "E2E_RC=$?"
tail -1 gpurun_out/bench_e2e_256.log
