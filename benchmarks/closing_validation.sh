cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
echo "=== gpu suite ==="
timeout 600 python -m pytest tests/ -q -m gpu 2>&1 | grep -E "FAILED|ERROR|passed|failed" | tail -2
echo "=== smoke ==="
timeout 240 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | grep -E "smoke|Error" | tail -1
echo "=== serial script ==="
MI355X_SYNTHETIC=1 MI355X_MODEL=net MI355X_EPOCHS=1 timeout 300 python cifar_example.py 2>&1 | grep -E "Accuracy|Error" | tail -1
echo "=== ddp script world-1 ==="
MI355X_SYNTHETIC=1 MI355X_EPOCHS=1 timeout 300 python -m torch.distributed.run --standalone --local-addr 127.0.0.1 --nproc-per-node 1 cifar_example_ddp.py 2>&1 | grep -E "Accuracy|Error" | tail -1
echo "=== r50 ==="
timeout 420 python bench.py --model resnet50 --size 224 --steps 8 --warmup 5 2>&1 | tail -1 | python -c "import json,sys; d=json.load(sys.stdin); print('r50', d['value'])"
echo "=== r18 ==="
timeout 240 python bench.py --steps 12 --warmup 6 2>&1 | tail -1 | python -c "import json,sys; d=json.load(sys.stdin); print('r18', d['value'])"
