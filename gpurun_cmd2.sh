set -x
cd /root/repo
python -m pytest tests/test_gpu_parity.py tests/test_gpu_slice.py tests/test_driver.py tests/test_decimal.py -m gpu -q > gpurun_out/r2_pytest2.log 2>&1
tail -3 gpurun_out/r2_pytest2.log
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline > gpurun_out/r2b_c2.json 2>/dev/null
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline > gpurun_out/r2b_c3.json 2>/dev/null
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r2b_prof_c3 -- python /root/repo/bench.py --workload c3 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2b_c3_prof.log 2>&1
timeout 400 rocprofv3 --pmc FETCH_SIZE --kernel-trace -d /root/repo/gpurun_out/r2b_pmc_f -- python /root/repo/bench.py --workload c3 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2b_pmc_f.log 2>&1
timeout 400 rocprofv3 --pmc WRITE_SIZE --kernel-trace -d /root/repo/gpurun_out/r2b_pmc_w -- python /root/repo/bench.py --workload c3 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2b_pmc_w.log 2>&1
echo DONE
