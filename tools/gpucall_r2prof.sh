set -x
cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/r2prof
# kernel-trace + stats of the default bench (the headline config)
timeout 1200 rocprofv3 --kernel-trace --stats -d gpurun_out/r2prof/kt -- \
  python bench.py --steps 3 --warmup 1 --skip-cpu-baseline > gpurun_out/r2prof/bench_kt.json 2> gpurun_out/r2prof/bench_kt.err
echo "kt rc=$?" | tee gpurun_out/r2prof/status.txt
# PMC passes (separate invocations, only --kernel-trace combined)
for pmc in "FETCH_SIZE" "WRITE_SIZE" "SQ_INSTS_VALU SQ_WAVE_CYCLES SQ_WAIT_ANY"; do
  name=$(echo $pmc | tr ' ' '_')
  timeout 900 rocprofv3 --kernel-trace --pmc $pmc -d gpurun_out/r2prof/pmc_$name -- \
    python bench.py --steps 2 --warmup 1 --batch 16384 --skip-cpu-baseline \
    > gpurun_out/r2prof/pmc_$name.json 2> gpurun_out/r2prof/pmc_$name.err
  echo "pmc $name rc=$?" | tee -a gpurun_out/r2prof/status.txt
done
ls -la gpurun_out/r2prof/ | head -20
