"""Summarize a rocprofv3 results.db into a small text table (run on the
GPU box so only the summary travels back).

  python tools/prof_summarize.py <results.db> [N]
"""
import sqlite3
import sys


def main():
  path = sys.argv[1]
  n = int(sys.argv[2]) if len(sys.argv) > 2 else 40
  db = sqlite3.connect(path)
  cur = db.cursor()
  tabs = [r[0] for r in cur.execute(
      "SELECT name FROM sqlite_master WHERE type='table' "
      "OR type='view'")]
  syms = [t for t in tabs if "kernel_symbol" in t]
  disps = [t for t in tabs if "kernel_dispatch" in t]
  if not syms or not disps:
    print("schema tables/views:", tabs)
    return
  sym = syms[0]
  disp = disps[0]
  q = (f"SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6 "
       f"FROM {disp} d JOIN {sym} s ON d.kernel_id=s.id "
       f"GROUP BY s.display_name ORDER BY 3 DESC LIMIT {n}")
  rows = list(cur.execute(q))
  total = sum(r[2] for r in rows)
  print(f"total kernel ms (top {n}): {total:.1f}")
  for name, cnt, ms in rows:
    print(f"{ms:9.3f} ms {cnt:6d}  {name[:100]}")


if __name__ == "__main__":
  main()
