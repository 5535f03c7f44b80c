cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0 MI355X_BACKEND=gloo
timeout 300 python -m torch.distributed.run --standalone --local-addr 127.0.0.1 --nproc-per-node 2 bench.py --gpus 2 --steps 8 --warmup 4 > gpurun_out/dp2.log 2>&1
echo "rc=$?"; grep -o '{"metric.*' gpurun_out/dp2.log | tail -1
