"""Extract kernel-trace / PMC summaries from rocprofv3 rocpd sqlite DBs into
markdown tables (committed under profiles/)."""
import sqlite3, sys, collections

def suffix(c):
    r = c.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'").fetchone()
    return r[0][len('rocpd_kernel_dispatch'):] if r else None

def kernel_table(db):
    c = sqlite3.connect(db)
    sfx = suffix(c)
    q = f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start),
                   MAX(ks.arch_vgpr_count), MAX(kd.private_segment_size),
                   MAX(kd.group_segment_size)
            FROM rocpd_kernel_dispatch{sfx} kd
            JOIN rocpd_info_kernel_symbol{sfx} ks ON kd.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY 3 DESC"""
    rows = c.execute(q).fetchall()
    tot = sum(r[2] for r in rows) or 1
    print("| kernel | calls | total ms | % | us/call | VGPR | scratch B | LDS B |")
    print("|---|---|---|---|---|---|---|---|")
    for name, n, ns, vgpr, scr, lds in rows[:14]:
        print(f"| `{name[:60]}` | {n} | {ns/1e6:.2f} | {100*ns/tot:.1f} | {ns/1e3/n:.1f} | {vgpr} | {scr} | {lds} |")

def pmc_table(db):
    c = sqlite3.connect(db)
    sfx = suffix(c)
    q = f"""SELECT ks.display_name, p.name, AVG(pe.value), COUNT(*)
            FROM rocpd_pmc_event{sfx} pe
            JOIN rocpd_info_pmc{sfx} p ON pe.pmc_id = p.id
            JOIN rocpd_kernel_dispatch{sfx} kd ON pe.event_id = kd.event_id
            JOIN rocpd_info_kernel_symbol{sfx} ks ON kd.kernel_id = ks.id
            GROUP BY ks.display_name, p.name"""
    try:
        rows = c.execute(q).fetchall()
    except sqlite3.OperationalError as e:
        # schema variant: pmc events keyed directly to dispatch
        q = f"""SELECT '-', p.name, AVG(pe.value), COUNT(*)
                FROM rocpd_pmc_event{sfx} pe
                JOIN rocpd_info_pmc{sfx} p ON pe.pmc_id = p.id
                GROUP BY p.name"""
        rows = c.execute(q).fetchall()
    by_kernel = collections.defaultdict(list)
    for kn, pn, avg, n in rows:
        by_kernel[kn].append((pn, avg, n))
    for kn, vals in sorted(by_kernel.items()):
        if 'k_render' in kn or 'k_wf' in kn or kn == '-':
            print(f"\n**{kn[:70]}**\n")
            print("| counter | n | avg/dispatch |")
            print("|---|---|---|")
            for pn, avg, n in sorted(vals):
                print(f"| {pn} | {n} | {avg:.4g} |")

if __name__ == "__main__":
    mode, db = sys.argv[1], sys.argv[2]
    (kernel_table if mode == "kt" else pmc_table)(db)
