set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3 > gpurun_out/r02_gputests3.log
timeout 900 python scripts/r02_convergence.py gpurun_out/r02_convergence.md > gpurun_out/r02_conv.log 2>&1
cd /tmp && export TMPDIR=/tmp && cd /root/repo
rm -rf gpurun_out/prof_gg3
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_gg3 -- python bench.py --model googlenet --steps 30 --warmup 13 > gpurun_out/r02_gg_prof3.log 2>&1
cat gpurun_out/r02_gputests3.log
tail -3 gpurun_out/r02_conv.log
grep metric gpurun_out/r02_gg_prof3.log
