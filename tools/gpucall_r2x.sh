set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
timeout 800 rocprofv3 --kernel-trace --stats -d gpurun_out/r2x_prof -- python bench.py --mode stream --steps 3 --warmup 1 > gpurun_out/r2x_stream.json 2> gpurun_out/r2x_stream.err
echo "prof=$?"
python3 - << 'PYEOF'
import sqlite3, glob, re
dbs = glob.glob('gpurun_out/r2x_prof/**/*_results.db', recursive=True)
rows = []
if dbs:
    c = sqlite3.connect(dbs[0])
    tabs = {r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")}
    kd = next(t for t in tabs if t.startswith('rocpd_kernel_dispatch'))
    ks = next(t for t in tabs if t.startswith('rocpd_info_kernel_symbol'))
    rows = list(c.execute(f"SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e6 FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id GROUP BY s.display_name ORDER BY 3 DESC LIMIT 14"))
with open('gpurun_out/r2x_kernels.txt', 'w') as f:
    for name, calls, tot, avg in rows:
        short = re.sub(r'\(.*', '', name.replace('void ', '')).strip().split('<')[0]
        line = f"{short:46s} {calls:5d} {tot:9.2f} {avg:8.3f}"
        print(line)
        f.write(line + "\n")
PYEOF
tail -c 300 gpurun_out/r2x_stream.json
