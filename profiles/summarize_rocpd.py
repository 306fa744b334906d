import sqlite3, glob, sys
out = []
for d in sys.argv[1:]:
    for db_path in sorted(glob.glob(f'gpurun_out/{d}/**/*_results.db', recursive=True) or glob.glob(f'{d}/**/*_results.db', recursive=True)):
        db = sqlite3.connect(db_path)
        cur = db.cursor()
        tabs = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
        u = [t for t in tabs if t.startswith('rocpd_kernel_dispatch')][0].replace('rocpd_kernel_dispatch_','')
        out.append(f"== {db_path} ==")
        q = f"""SELECT ks.kernel_name, COUNT(*), SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e6,
               MAX(ks.arch_vgpr_count), MAX(ks.private_segment_size)
               FROM rocpd_kernel_dispatch_{u} k JOIN rocpd_info_kernel_symbol_{u} ks ON k.kernel_id=ks.id
               GROUP BY ks.kernel_name ORDER BY 3 DESC"""
        out.append("%-26s %5s %10s %9s %5s %8s" % ("kernel","n","total_ms","avg_ms","vgpr","scratch"))
        for r in cur.execute(q):
            out.append("%-26s %5d %10.2f %9.3f %5d %8d" % (r[0][:26],r[1],r[2],r[3],r[4],r[5]))
        try:
            q2 = f"""SELECT ks.kernel_name, pi.name, COUNT(*), AVG(pe.value)
                 FROM rocpd_pmc_event_{u} pe JOIN rocpd_kernel_dispatch_{u} k ON pe.event_id=k.event_id
                 JOIN rocpd_info_kernel_symbol_{u} ks ON k.kernel_id=ks.id
                 JOIN rocpd_info_pmc_{u} pi ON pe.pmc_id=pi.id
                 GROUP BY ks.kernel_name, pi.name ORDER BY 4 DESC LIMIT 10"""
            rows = list(cur.execute(q2))
            if rows:
                out.append("-- PMC (value unit: KB for FETCH/WRITE_SIZE) --")
                for r in rows:
                    out.append("%-26s %-12s n=%3d avg=%14.0f" % (r[0][:26],r[1],r[2],r[3]))
        except Exception:
            pass
print("\n".join(out))
