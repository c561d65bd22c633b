cd /root/repo
python -m pytest tests/test_gpu_parity.py -m gpu -q 2>&1 | tail -2
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline 2>gpurun_out/e1 | python -c "import json,sys; d=json.load(sys.stdin); print('C2', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')"
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline > gpurun_out/r2c_c3.json 2>gpurun_out/e2
python -c "import json; d=json.load(open('gpurun_out/r2c_c3.json')); print('C3', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')"
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c_prof_c3 -- python /root/repo/bench.py --workload c3 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2c_prof.log 2>&1
grep "k_probe" /root/repo/gpurun_out/r2c_prof_c3/runc/*kernel_stats.csv
find /root/repo/gpurun_out -name "*.db" -delete
echo DONE
