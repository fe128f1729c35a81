set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -x -q 2>&1 | tail -5 > gpurun_out/r02_gputests.log
echo "=== bench alexnet bf16 ===" >> gpurun_out/r02_bench.log
timeout 600 python bench.py --steps 50 --warmup 10 >> gpurun_out/r02_bench.log 2>&1
echo "=== bench alexnet fp32 ===" >> gpurun_out/r02_bench.log
timeout 600 python bench.py --steps 50 --warmup 10 --dtype fp32 >> gpurun_out/r02_bench.log 2>&1
echo "=== bench googlenet bf16 ===" >> gpurun_out/r02_bench.log
timeout 600 python bench.py --model googlenet --steps 50 --warmup 10 >> gpurun_out/r02_bench.log 2>&1
echo "=== bench vgg16 bf16 ===" >> gpurun_out/r02_bench.log
timeout 600 python bench.py --model vgg16 --steps 30 --warmup 5 >> gpurun_out/r02_bench.log 2>&1
echo "=== 2-rank gloo-cuda graph-fallback ===" >> gpurun_out/r02_bench.log
PS_BACKEND=gloo timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29518 bench.py --gpus 2 --steps 5 --warmup 2 --batch 32 >> gpurun_out/r02_bench.log 2>&1
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_gg -- python bench.py --model googlenet --steps 20 --warmup 5 --no-graph > gpurun_out/r02_gg_prof.log 2>&1
tail -3 gpurun_out/r02_gputests.log
grep -h '"metric"' gpurun_out/r02_bench.log
