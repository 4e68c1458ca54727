set -x
cd /root/repo
mkdir -p gpurun_out
python -c "import torch; print(torch.__version__, torch.cuda.get_device_name(0))" 2>&1
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu_r2a.log 2>&1
echo "PYTEST_RC=$?"
tail -5 gpurun_out/pytest_gpu_r2a.log
# small e2e smoke: fast failure detection
timeout 420 python bench.py --steps 1 --warmup 0 --concurrency 32 --turns 2 --rag-docs 2000 > gpurun_out/bench_e2e_smoke.log 2>&1
echo "SMOKE_RC=$?"
tail -2 gpurun_out/bench_e2e_smoke.log
# full headline config
timeout 1500 python bench.py --steps 2 --warmup 1 > gpurun_out/bench_e2e_full.log 2>&1
echo "FULL_RC=$?"
tail -2 gpurun_out/bench_e2e_full.log
