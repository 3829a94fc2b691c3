"""Summarize a rocprofv3 counter_collection.csv per kernel (per-dispatch avg)."""
import csv, collections, glob, sys

pat = sys.argv[1] if len(sys.argv) > 1 else "/tmp/pmc/*counter_collection.csv"
files = glob.glob(pat)
if not files:
    print("no counter csv at", pat); sys.exit(1)
agg = collections.defaultdict(lambda: collections.defaultdict(float))
cnt = collections.Counter()
for fn in files:
    for r in csv.DictReader(open(fn)):
        name = r["Kernel_Name"]
        name = name.replace("(anonymous namespace)::", "")
        k = name.split("(")[0].split("<")[0].replace("void ", "").strip()[:60]
        agg[k][r["Counter_Name"]] += float(r["Counter_Value"])
        cnt[k] = cnt[k]
        cnt[(k, r["Counter_Name"])] += 1
keys = sys.argv[2].split(",") if len(sys.argv) > 2 else ["fused_edge", "tall_linear", "wgrad", "seg_reduce"]
for k, d in agg.items():
    if not any(s in k for s in keys):
        continue
    n = max(1, cnt[(k, "SQ_WAVE_CYCLES")]) if (k, "SQ_WAVE_CYCLES") in cnt else max(
        1, max(cnt[(k, c)] for c in d))
    print(f"== {k}  dispatches={n}")
    for c, v in sorted(d.items()):
        print(f"   {c:26s} {v/n:16.0f}")
