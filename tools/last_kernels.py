import glob, sqlite3, sys
db = glob.glob(sys.argv[1])[0]
con = sqlite3.connect(db)
cur = con.cursor()
sfx = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch_","")
rows = list(cur.execute(f"""
    SELECT kd.start, ks.display_name FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
    ORDER BY kd.start DESC LIMIT 12"""))
for t, n in rows[::-1]:
    print(n[:110])
