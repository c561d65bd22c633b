cd /root/repo
python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 600 python bench.py --workload c4 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e3 | python -c "import json,sys; d=json.load(sys.stdin); print('C4', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -2 gpurun_out/e3
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e2 | python -c "import json,sys; d=json.load(sys.stdin); print('C3', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms')" || tail -2 gpurun_out/e2
echo DONE
