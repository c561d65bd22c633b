cd /root/repo
python -m pytest tests/test_gpu_fuzz.py tests/test_range_frames.py tests/test_q18_pipeline.py tests/test_q9_pipeline.py -m gpu -q 2>&1 | tail -2
timeout 300 python -m pytest tests/test_rccl_smoke.py -m gpu -q -rs 2>&1 | tail -4
timeout 600 python bench.py --workload c4 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e3 | python -c "import json,sys; d=json.load(sys.stdin); print('C4', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -2 gpurun_out/e3
timeout 600 python bench.py --workload c5 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e4 | python -c "import json,sys; d=json.load(sys.stdin); print('C5', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -2 gpurun_out/e4
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --pmc FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2g_pmc_c2f -- python /root/repo/bench.py --workload c2 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2g1.log 2>&1
timeout 500 rocprofv3 --pmc WRITE_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2g_pmc_c2w -- python /root/repo/bench.py --workload c2 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2g2.log 2>&1
timeout 500 rocprofv3 --pmc FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2g_pmc_c4f -- python /root/repo/bench.py --workload c4 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2g3.log 2>&1
timeout 500 rocprofv3 --pmc WRITE_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2g_pmc_c4w -- python /root/repo/bench.py --workload c4 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2g4.log 2>&1
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2g_prof_c4 -- python /root/repo/bench.py --workload c4 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2g5.log 2>&1
find /root/repo/gpurun_out -name "*.db" -delete
echo DONE
