cd /root/repo
python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 900 python bench.py --workload c3 --steps 10 --warmup 2 --no-cpu-baseline 2>gpurun_out/e2 | python -c "import json,sys; d=json.load(sys.stdin); print('C3 x10 steps', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms/step, frac', round(d['roofline']['frac'],3))" || tail -2 gpurun_out/e2
echo DONE
