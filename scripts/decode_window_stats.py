import sqlite3, glob, sys
db = sorted(glob.glob(sys.argv[1] + '/**/*.db', recursive=True))[0]
con = sqlite3.connect(db)
tabs = [r[0] for r in con.execute(
    "select name from sqlite_master where type in ('table','view')")]
kd = [t for t in tabs if 'kernel_dispatch' in t and t.startswith('rocpd')][0]
ks = [t for t in tabs if 'info_kernel_symbol' in t][0]
t0 = con.execute(
    f"select max(kd.end) from {kd} kd join {ks} ks on kd.kernel_id=ks.id "
    f"where ks.display_name like '%paged_prefill%'").fetchone()[0] or 0
rows = con.execute(
    f"select ks.display_name, count(*), sum(kd.end-kd.start)/1e6, "
    f"avg(kd.end-kd.start)/1e3 from {kd} kd join {ks} ks "
    f"on kd.kernel_id=ks.id where kd.start > ? "
    f"group by 1 order by 3 desc limit 22", (t0,)).fetchall()
tot = sum(r[2] for r in rows)
span = con.execute(f"select (max(end)-min(start))/1e6 from {kd} "
                   f"where start > ?", (t0,)).fetchone()[0]
print(f"decode window: {span:.1f} ms wall, {tot:.1f} ms kernel-sum")
for n, c, s, a in rows:
    print(f"{s:8.1f} ms  n={c:<6} avg={a:7.1f} us  {n[:90]}")
