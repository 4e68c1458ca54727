set -x
cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/prof_r2
timeout 360 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r2 -- python bench.py --mode engine --concurrency 256 --steps 1 --warmup 1 > gpurun_out/prof_r2/bench.log 2>&1
echo "PROF_RC=$?"
ls gpurun_out/prof_r2/ | head
tail -1 gpurun_out/prof_r2/bench.log
