import csv
import glob
import sys

f = glob.glob("/tmp/tr/**/*kernel_trace*.csv", recursive=True)[0]
rows = list(csv.DictReader(open(f)))
rows.sort(key=lambda r: int(r["Start_Timestamp"]))


def dur(r):
    return (int(r["End_Timestamp"]) - int(r["Start_Timestamp"])) / 1e6


big = [r for r in rows if "elementwise" in r["Kernel_Name"]
       and dur(r) > 0.3]
print(f"{len(big)} copies > 0.3 ms; total {sum(dur(r) for r in big):.1f} ms")
for r in big[-40:]:
    i = rows.index(r)
    prev = rows[i - 1]["Kernel_Name"][:36] if i else ""
    nxt = rows[i + 1]["Kernel_Name"][:36] if i + 1 < len(rows) else ""
    print(f"{dur(r):6.2f} ms  after={prev!r:40} before={nxt!r}")
