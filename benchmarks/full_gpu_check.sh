cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 600 python -m pytest tests/ -q -m gpu 2>&1 | grep -E "FAILED|ERROR|passed|failed" | tail -3
timeout 240 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -2
MI355X_SYNTHETIC=1 MI355X_MODEL=resnet18 MI355X_EPOCHS=1 timeout 300 python cifar_example.py 2>&1 | tail -1
