cd /root/repo
python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline 2>gpurun_out/e1 | python -c "import json,sys; d=json.load(sys.stdin); print('C2', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3))" || tail -3 gpurun_out/e1
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e2 | python -c "import json,sys; d=json.load(sys.stdin); print('C3', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3))" || tail -3 gpurun_out/e2
timeout 600 python bench.py --workload c4 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e3 | python -c "import json,sys; d=json.load(sys.stdin); print('C4', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -3 gpurun_out/e3
timeout 600 python bench.py --workload c5 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e4 | python -c "import json,sys; d=json.load(sys.stdin); print('C5', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -3 gpurun_out/e4
timeout 300 python bench.py --workload c2chunk --steps 2 --warmup 1 --no-cpu-baseline 2>gpurun_out/e5 | python -c "import json,sys; d=json.load(sys.stdin); print('C2chunk(sf1,1000)', round(d['value']/1e9,3),'G/s', round(d['ms_per_step'],2),'ms/step')" || tail -3 gpurun_out/e5
echo DONE
