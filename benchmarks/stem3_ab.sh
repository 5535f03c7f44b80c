set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 240 python -m pytest tests/test_gpu_kernels.py -q -m gpu -k "conv2d_fwd_bwd or stem" -x 2>&1 | grep -E "FAILED|ERROR|passed|failed" | tail -3
# interleaved A/B: new stem3 GEMM (default) vs old dot2 LDS stem (GEMM off)
for rep in 1 2; do
  timeout 240 python bench.py --steps 12 --warmup 6 2>&1 | tail -1
  MI355X_STEM_GEMM=0 timeout 240 python bench.py --steps 12 --warmup 6 2>&1 | tail -1
done
