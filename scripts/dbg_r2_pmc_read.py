import sqlite3, glob, collections, sys
db = glob.glob(sys.argv[1])[0]
c = sqlite3.connect(db)
suf = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")
       if r[0].startswith('rocpd_metadata')][0].replace('rocpd_metadata', '')
q = (f"SELECT substr(ks.display_name,1,42), ip.name, SUM(p.value) "
     f"FROM rocpd_pmc_event{suf} p "
     f"JOIN rocpd_kernel_dispatch{suf} kd ON p.event_id=kd.event_id "
     f"JOIN rocpd_info_kernel_symbol{suf} ks ON kd.kernel_id=ks.id "
     f"JOIN rocpd_info_pmc{suf} ip ON p.pmc_id=ip.id "
     f"WHERE ks.display_name NOT LIKE '%at::%' GROUP BY 1,2 ORDER BY 1")
agg = collections.defaultdict(dict)
for n, cn, v in c.execute(q):
    agg[n][cn] = v
for n, d in agg.items():
    cf = d.get('SQ_LDS_BANK_CONFLICT', 0)
    bz = d.get('SQ_BUSY_CYCLES', 0)
    print(f"{n:44s} conflicts={cf:.2e} busy={bz:.2e}")
