cd /root/repo
python -m pytest tests/test_gpu_parity.py tests/test_gpu_fuzz.py -m gpu -q 2>&1 | tail -2
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline 2>gpurun_out/e1 | python -c "import json,sys; d=json.load(sys.stdin); print('C2 radix-auto', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -3 gpurun_out/e1
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline > gpurun_out/r2d_c3.json 2>gpurun_out/e2 && python -c "import json; d=json.load(open('gpurun_out/r2d_c3.json')); print('C3 radix-auto', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -3 gpurun_out/e2
GX_RADIX=0 timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e3 | python -c "import json,sys; d=json.load(sys.stdin); print('C3 radix-off ', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -3 gpurun_out/e3
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2d_prof_c3 -- python /root/repo/bench.py --workload c3 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2d_prof.log 2>&1
grep -E "k_probe|k_radix" /root/repo/gpurun_out/r2d_prof_c3/runc/*kernel_stats.csv
find /root/repo/gpurun_out -name "*.db" -delete
echo DONE
