cd /root/repo
timeout 300 python bench.py --workload c2 --steps 5 --warmup 2 --no-cpu-baseline > gpurun_out/r2b_c2.json 2> gpurun_out/r2b_c2.err
echo "C2 rc=$?"; cat gpurun_out/r2b_c2.json; tail -3 gpurun_out/r2b_c2.err
timeout 600 python bench.py --workload c3 --steps 3 --warmup 1 --no-cpu-baseline > gpurun_out/r2b_c3.json 2> gpurun_out/r2b_c3.err
echo "C3 rc=$?"; cat gpurun_out/r2b_c3.json; tail -3 gpurun_out/r2b_c3.err
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2b_prof_c3 -- python /root/repo/bench.py --workload c3 --steps 2 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2b_c3_prof.log 2>&1
timeout 400 rocprofv3 --pmc FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2b_pmc_f -- python /root/repo/bench.py --workload c3 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2b_pmc_f.log 2>&1
timeout 400 rocprofv3 --pmc WRITE_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2b_pmc_w -- python /root/repo/bench.py --workload c3 --steps 1 --warmup 1 --no-cpu-baseline > /root/repo/gpurun_out/r2b_pmc_w.log 2>&1
find /root/repo/gpurun_out -name "*.db" -delete
du -sh /root/repo/gpurun_out
grep -h "k_probe" /root/repo/gpurun_out/r2b_prof_c3/*kernel_stats.csv 2>/dev/null | head -3
echo DONE
