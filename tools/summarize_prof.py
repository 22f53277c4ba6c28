import sqlite3, glob, sys
db = sorted(glob.glob(sys.argv[1] + "/**/*.db", recursive=True))[0]
con = sqlite3.connect(db)
sfx = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch_", "")
q = f"""SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e3
FROM rocpd_kernel_dispatch_{sfx} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30"""
tot = list(con.execute(f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{sfx}"))[0][0]
print(f"TOTAL GPU ms: {tot:.1f}")
for name, n, ms, avg in con.execute(q):
    print(f"{name.split('(')[0][:70]:<72} {n:>6} {ms:>9.1f} {avg:>8.1f} {100*ms/tot:5.1f}%")
