set -x
cd /tmp && export TMPDIR=/tmp
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/prof_stem3
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_stem3 -o s3 -- python bench.py --steps 7 --warmup 5 > gpurun_out/prof_stem3/bench.log 2>&1
tail -1 gpurun_out/prof_stem3/bench.log
ls gpurun_out/prof_stem3/ | head
MI355X_FP16=1 timeout 240 python bench.py --fp16 --steps 12 --warmup 6 2>&1 | tail -1
