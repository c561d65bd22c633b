cd /root/repo
for W in c2 c4 c5; do
timeout 900 python bench.py --workload $W --steps 5 --warmup 2 > gpurun_out/r2z_$W.json 2>gpurun_out/r2z_$W.err && python -c "
import json; d=json.load(open('gpurun_out/r2z_$W.json'))
cb=d['cpu_baseline']
print('$W', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms | cpu', round(cb['value']/1e6,1),'M/s x',cb['cores'],'cores')" || tail -3 gpurun_out/r2z_$W.err
done
timeout 900 python bench.py > gpurun_out/r2z_default.json 2>gpurun_out/r2z_d.err && python -c "
import json; d=json.load(open('gpurun_out/r2z_default.json'))
print('default(c3)', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3), '| traffic', d['roofline']['traffic'])" || tail -3 gpurun_out/r2z_d.err
tools/gx_driver --lib galaxysql_amd/csrc/libgxhip.so --device 0 bench-agg --probe-rows 600000000 --build-rows 150000000 --steps 2 2>&1 | tail -1
echo DONE
