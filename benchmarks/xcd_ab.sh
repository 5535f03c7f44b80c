cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
MI355X_XCD_SWZ=1 timeout 240 python -m pytest tests/test_gpu_kernels.py -q -m gpu -k "conv2d_fwd_bwd" -x 2>&1 | grep -E "FAILED|ERROR|passed|failed" | tail -2
for rep in 1 2; do
  MI355X_XCD_SWZ=1 timeout 240 python bench.py --steps 12 --warmup 6 2>&1 | tail -1 | python -c "import json,sys; d=json.load(sys.stdin); print('SWZ1', d['value'])"
  timeout 240 python bench.py --steps 12 --warmup 6 2>&1 | tail -1 | python -c "import json,sys; d=json.load(sys.stdin); print('SWZ0', d['value'])"
done
