cd /root/repo
python -m pytest tests -m gpu -q 2>&1 | tail -2
python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" 2>&1 | tail -1
tools/gx_driver --lib galaxysql_amd/csrc/libgxhip.so --device 0 selftest 2>&1 | tail -3
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 2>gpurun_out/e2 | python -c "import json,sys; d=json.load(sys.stdin); print('C3 full', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3), 'cpu', round(d['cpu_baseline']['value']/1e6,1),'M rows/s x', d['cpu_baseline']['cores'],'cores')" || tail -3 gpurun_out/e2
echo DONE
