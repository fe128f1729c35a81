set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -4 > gpurun_out/r02_gputests2.log
echo "=== bench googlenet bf16 (graphed) ===" > gpurun_out/r02_bench2.log
timeout 300 python bench.py --model googlenet --steps 100 --warmup 10 >> gpurun_out/r02_bench2.log 2>&1
echo "=== bench alexnet bf16 (graphed) ===" >> gpurun_out/r02_bench2.log
timeout 300 python bench.py --steps 100 --warmup 10 >> gpurun_out/r02_bench2.log 2>&1
echo "=== bench vgg16 ===" >> gpurun_out/r02_bench2.log
timeout 300 python bench.py --model vgg16 --steps 50 --warmup 5 >> gpurun_out/r02_bench2.log 2>&1
cd /tmp && export TMPDIR=/tmp && cd /root/repo
rm -rf gpurun_out/prof_gg2 && timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_gg2 -- python bench.py --model googlenet --steps 20 --warmup 5 > gpurun_out/r02_gg_prof2.log 2>&1
cat gpurun_out/r02_gputests2.log
grep -h '"metric"' gpurun_out/r02_bench2.log gpurun_out/r02_gg_prof2.log
