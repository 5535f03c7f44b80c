cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
for rep in 1 2 3; do
  timeout 240 python bench.py --steps 10 --warmup 5 2>&1 | tail -1 | python -c "import json,sys; print('GEMM1', json.load(sys.stdin)['value'])"
  MI355X_STEM_GEMM=0 timeout 240 python bench.py --steps 10 --warmup 5 2>&1 | tail -1 | python -c "import json,sys; print('GEMM0', json.load(sys.stdin)['value'])"
done
