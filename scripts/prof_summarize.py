"""Summarize a rocprofv3 results.db into a small top-kernels text table."""
import sys, sqlite3, glob
db = sorted(glob.glob(sys.argv[1] + "/**/*results.db", recursive=True))[-1]
cur = sqlite3.connect(db).cursor()
print(f"{'kernel':<60} {'calls':>6} {'total_ms':>10} {'avg_us':>8} {'pct':>6}")
for name, calls, tot, avg, pct in cur.execute(
        "SELECT name, total_calls, total_duration, average, percentage "
        "FROM top_kernels LIMIT 20"):
    clean = name.replace("(anonymous namespace)::", "").replace("void ", "")
    short = clean.split("(")[0][:58]
    print(f"{short:<60} {calls:>6} {tot/1000:>10.2f} {avg:>8.1f} {pct:>6.2f}")
