cd /root/repo
timeout 300 python -m pytest tests/test_gpu_fusion.py -x -q > gpurun_out/pyf.log 2>&1; echo "PYTEST rc=$?"; tail -1 gpurun_out/pyf.log
for env in "BS_FUSE_NOSWIZ=1" ""; do
  timeout 300 env $env python tools/bench_fusion.py --steps 3 --warmup 1 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print(\"[$env]\", round(d[\"value\"]/1e9,2), \"Gvox/s kernel\", d[\"kernel_achieved_GBs\"], \"GB/s\")"
done
export TMPDIR=/tmp; cd /tmp
timeout 400 rocprofv3 --pmc FETCH_SIZE -d /root/repo/gpurun_out/prof -- python /root/repo/tools/bench_fusion.py --steps 1 --warmup 1 > /root/repo/gpurun_out/prof_fuse.log 2>&1
echo "PMC rc=$?"
python - <<'PYEOF'
import glob, csv, collections
f = sorted(glob.glob("/root/repo/gpurun_out/prof/**/*.csv", recursive=True))
print(f[:5])
agg = collections.defaultdict(float); cnt = collections.defaultdict(int)
for fn in f:
    for r in csv.DictReader(open(fn)):
        n = r.get("Kernel_Name", r.get("kernel_name", ""))[:40]
        v = r.get("Counter_Value", r.get("FETCH_SIZE", 0))
        try: agg[n] += float(v); cnt[n] += 1
        except: pass
for k in sorted(agg, key=lambda k: -agg[k])[:6]:
    print(k, cnt[k], round(agg[k]/max(1,cnt[k]),1))
PYEOF
