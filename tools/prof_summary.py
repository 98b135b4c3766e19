"""Summarize a rocprofv3 rocpd .db: total time per kernel name."""
import glob
import sys
import sqlite3


def main(path):
    dbs = glob.glob(path) if '*' in path else [path]
    con = sqlite3.connect(dbs[0])
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith('rocpd_kernel_dispatch'))
    ks = next(t for t in tables if t.startswith('rocpd_info_kernel_symbol'))
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6 as ms
    FROM {kd} kd JOIN {ks} s ON kd.kernel_id = s.id
    GROUP BY s.display_name ORDER BY ms DESC LIMIT 40
    """
    rows = con.execute(q).fetchall()
    tot = sum(r[2] for r in rows)
    for name, cnt, ms in rows:
        print(f'{ms:10.1f} ms {cnt:6d}  {name[:110]}')
    print(f'TOTAL {tot:.1f} ms')


if __name__ == '__main__':
    main(sys.argv[1])
