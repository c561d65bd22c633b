cd /root/repo
python -m pytest tests/test_gpu_parity.py tests/test_gpu_fuzz.py tests/test_c1_plumbing.py -m gpu -q 2>&1 | tail -2
for B in 0 1; do
GX_BLOOM=$B timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline 2>gpurun_out/e1 | python -c "import json,sys; d=json.load(sys.stdin); print('C2 bloom=$B', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -2 gpurun_out/e1
GX_BLOOM=$B timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e2 | python -c "import json,sys; d=json.load(sys.stdin); print('C3 bloom=$B', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -2 gpurun_out/e2
done
timeout 300 python bench.py --workload c2chunk --steps 2 --warmup 1 --flush-rows 1048576 --no-cpu-baseline 2>gpurun_out/e5 | python -c "import json,sys; d=json.load(sys.stdin); print('C2chunk push/flush(1M)', round(d['value']/1e9,3),'G/s', round(d['ms_per_step'],2),'ms/step')" || tail -3 gpurun_out/e5
timeout 300 python bench.py --workload c2chunk --steps 2 --warmup 1 --flush-rows 65536 --no-cpu-baseline 2>gpurun_out/e6 | python -c "import json,sys; d=json.load(sys.stdin); print('C2chunk push/flush(64K)', round(d['value']/1e9,3),'G/s', round(d['ms_per_step'],2),'ms/step')" || tail -3 gpurun_out/e6
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2e_prof_c5 -- python /root/repo/bench.py --workload c5 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2e_c5_prof.log 2>&1
tail -1 /root/repo/gpurun_out/r2e_c5_prof.log
head -12 /root/repo/gpurun_out/r2e_prof_c5/runc/*kernel_stats.csv | cut -c1-110
find /root/repo/gpurun_out -name "*.db" -delete
echo DONE
