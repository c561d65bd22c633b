cd /root/repo
python -m pytest tests -m gpu -q 2>&1 | tail -2
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline 2>gpurun_out/e1 | python -c "import json,sys; d=json.load(sys.stdin); print('C2 loadcap', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3), 'ev', round(d['config']['probe_event_ms_per_launch'],2))" || tail -2 gpurun_out/e1
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e2 | python -c "import json,sys; d=json.load(sys.stdin); print('C3 loadcap', round(d['value']/1e9,2),'G/s', round(d['ms_per_step'],2),'ms, frac', round(d['roofline']['frac'],3))" || tail -2 gpurun_out/e2
timeout 300 python bench.py --workload c2chunk --chunk-size 65536 --flush-rows 1048576 --steps 3 --warmup 1 --no-cpu-baseline 2>gpurun_out/e5 | python -c "import json,sys; d=json.load(sys.stdin); print('C2chunk(cs=64K,flush=1M)', round(d['value']/1e9,3),'G/s')" || tail -3 gpurun_out/e5
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --pmc FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2h_pmc_c2f -- python /root/repo/bench.py --workload c2 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2h1.log 2>&1
timeout 400 rocprofv3 --pmc WRITE_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2h_pmc_c2w -- python /root/repo/bench.py --workload c2 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2h2.log 2>&1
python - <<'PYEOF'
import csv, glob
for tag in ("r2h_pmc_c2f", "r2h_pmc_c2w"):
    for f in glob.glob(f"/root/repo/gpurun_out/{tag}/runc/*counter_collection.csv"):
        for r in csv.DictReader(open(f)):
            if "k_probe" in r["Kernel_Name"]:
                print(tag, "k_probe", round(float(r["Counter_Value"])/1e6, 2), "GB")
PYEOF
find /root/repo/gpurun_out -name "*.db" -delete
echo DONE
