set -x
cd /root/repo
mkdir -p gpurun_out
for items in 16 8; do
  HBLS_COOP_ITEMS=$items timeout 420 python tools/coop_ab_child.py >> gpurun_out/r2g_coopab.log 2>&1
  echo "coop$items rc=$?" | tee -a gpurun_out/r2g_status.txt
done
# also A/B the stream bench between coop-items settings (ticks are 4096)
HBLS_COOP_ITEMS=16 timeout 500 python bench.py --mode stream --steps 3 --warmup 1 > gpurun_out/r2g_stream16.json 2>/dev/null
echo "stream16 rc=$?" | tee -a gpurun_out/r2g_status.txt
HBLS_COOP_ITEMS=8 timeout 500 python bench.py --mode stream --steps 3 --warmup 1 > gpurun_out/r2g_stream8.json 2>/dev/null
echo "stream8 rc=$?" | tee -a gpurun_out/r2g_status.txt
cat gpurun_out/r2g_coopab.log
python -c "import json; a=json.load(open('gpurun_out/r2g_stream16.json')); b=json.load(open('gpurun_out/r2g_stream8.json')); print('stream16', a['value'], 'stream8', b['value'])" 2>/dev/null
