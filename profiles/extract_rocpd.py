"""Extract kernel stats / PMC values from rocprofv3 rocpd SQLite outputs
(ROCm 7.2 default output format) into the committed summaries under
profiles/. Usage: python profiles/extract_rocpd.py <results.db> [pmc]"""
import sqlite3
import sys


def suffix(con, base):
    rows = [r[0] for r in con.execute(
        "select name from sqlite_master where type='table' and name like ?",
        (base + "%",))]
    return rows[0].replace(base + "_", "")


def kernel_stats(db):
    con = sqlite3.connect(db)
    suf = suffix(con, "rocpd_kernel_dispatch")
    return con.execute(f"""
      select k.display_name, count(*), sum(d.end-d.start)/1e6,
             avg(d.end-d.start)/1e6
      from rocpd_kernel_dispatch_{suf} d
      join rocpd_info_kernel_symbol_{suf} k on d.kernel_id=k.id
      group by 1 order by 3 desc""").fetchall()


def pmc_stats(db):
    con = sqlite3.connect(db)
    suf = suffix(con, "rocpd_kernel_dispatch")
    name = con.execute(
        f"select name from rocpd_info_pmc_{suf} limit 1").fetchall()
    rows = con.execute(f"""
      select k.display_name, count(*), avg(p.value), avg(d.end-d.start)/1e6
      from rocpd_pmc_event_{suf} p
      join rocpd_kernel_dispatch_{suf} d on p.event_id=d.event_id
      join rocpd_info_kernel_symbol_{suf} k on d.kernel_id=k.id
      group by 1 order by 3 desc""").fetchall()
    return name, rows


if __name__ == "__main__":
    db = sys.argv[1]
    if len(sys.argv) > 2 and sys.argv[2] == "pmc":
        name, rows = pmc_stats(db)
        print("counter:", name)
        for r in rows:
            print("%-40s n=%3d avg=%16.1f dur=%9.3f ms" % (r[0][:40], *r[1:]))
    else:
        for r in kernel_stats(db):
            print("%-40s n=%3d total=%10.3f ms avg=%9.3f ms"
                  % (r[0][:40], *r[1:]))
