cd /root/repo
python -m pytest tests/test_gpu_parity.py -m gpu -q 2>&1 | tail -1
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline > gpurun_out/r2f_c3.json 2>gpurun_out/e2 && python -c "
import json; d=json.load(open('gpurun_out/r2f_c3.json'))
print('C3 default(bloom)', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3), 'launches', d['config']['probe_launches'], 'ev_ms', round(d['config']['probe_event_ms_per_launch'],2))" || tail -2 gpurun_out/e2
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2f_prof_c3 -- python /root/repo/bench.py --workload c3 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2f_prof.log 2>&1
grep -E "k_probe|k_scan|k_agg_insert|k_gather_multi|k_bloom" /root/repo/gpurun_out/r2f_prof_c3/runc/*kernel_stats.csv | cut -c1-100
timeout 500 rocprofv3 --pmc FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2f_pmc_f -- python /root/repo/bench.py --workload c3 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2f_pmc_f.log 2>&1
timeout 500 rocprofv3 --pmc WRITE_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2f_pmc_w -- python /root/repo/bench.py --workload c3 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2f_pmc_w.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2f_prof_c2 -- python /root/repo/bench.py --workload c2 --steps 3 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2f_prof_c2.log 2>&1
grep -E "k_probe|k_gather_multi" /root/repo/gpurun_out/r2f_prof_c2/runc/*kernel_stats.csv | cut -c1-100
find /root/repo/gpurun_out -name "*.db" -delete
echo DONE
