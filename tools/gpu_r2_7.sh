cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/prof_r2c
timeout 150 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r2c -- python bench.py --mode engine --concurrency 64 --steps 1 --warmup 1 > gpurun_out/prof_r2c/bench.log 2>&1
echo "PROF_RC=$?"
find gpurun_out/prof_r2c -type f \( -name "*.db" -o -name "*kernel_trace*" \) -delete
find gpurun_out/prof_r2c -type f
du -sh gpurun_out/prof_r2c
