import sqlite3, glob, sys, csv, re
for f in sorted(glob.glob(sys.argv[1])):
    con = sqlite3.connect(f)
    tabs = [r[0] for r in con.execute("select name from sqlite_master where type='table'")]
    kd = [t for t in tabs if t.startswith('rocpd_kernel_dispatch')]
    if not kd: continue
    suf = kd[0][len('rocpd_kernel_dispatch'):]
    out = f.replace('_results.db', '_kernel_stats.csv')
    with open(out, 'w', newline='') as fh:
        w = csv.writer(fh)
        w.writerow(['Name','Calls','TotalDurationNs','AverageNs','Percentage'])
        rows = list(con.execute(f"""
            select s.display_name, count(*), sum(d.end-d.start), avg(d.end-d.start)
            from rocpd_kernel_dispatch{suf} d
            join rocpd_info_kernel_symbol{suf} s on d.kernel_id = s.id
            group by s.display_name order by 3 desc"""))
        tot = sum(r[2] for r in rows) or 1
        for name, n, t, a in rows:
            w.writerow([name, n, t, round(a,1), round(100*t/tot,2)])
    print('wrote', out)
