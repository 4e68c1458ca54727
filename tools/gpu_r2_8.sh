cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/prof_r2d
timeout 90 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof_r2d -- python bench.py --mode engine --concurrency 16 --steps 1 --warmup 0 > gpurun_out/prof_r2d/bench.log 2>&1
echo "PROF_RC=$?"
find gpurun_out/prof_r2d -type f ! -name "*stats*" ! -name "bench.log" -delete
find gpurun_out/prof_r2d -type f
