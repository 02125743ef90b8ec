"""Digest a rocprofv3 rocpd db into a small markdown table (run on the box)."""
import sqlite3, sys
db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
rows = list(cur.execute("SELECT name,total_calls,total_duration,average,percentage FROM top_kernels ORDER BY total_duration DESC LIMIT 16"))
with open(sys.argv[2], "w") as f:
    f.write("| kernel | calls | total ms | avg us | % GPU |\n|---|---|---|---|---|\n")
    for name, c, tot, avg, pct in rows:
        n = name.split("(")[0]
        if "at::native" in n:
            n = "torch " + n.split("<")[0].split("::")[-1]
        f.write(f"| `{n}` | {c} | {tot/1000:.1f} | {avg:.1f} | {pct:.1f} |\n")
print(open(sys.argv[2]).read())
