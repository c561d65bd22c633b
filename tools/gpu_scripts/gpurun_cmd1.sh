set -x
cd /root/repo
python -m pytest tests -m gpu -q > gpurun_out/r2_pytest1.log 2>&1
tail -5 gpurun_out/r2_pytest1.log
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --cpu-probe-rows 2000000 > gpurun_out/r2_b_c2.json 2> gpurun_out/r2_b_c2.err
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 > gpurun_out/r2_b_c3.json 2> gpurun_out/r2_b_c3.err
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r2_prof_c3 -- python /root/repo/bench.py --workload c3 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2_b_c3_prof.log 2>&1
tail -2 /root/repo/gpurun_out/r2_b_c3_prof.log
echo DONE
