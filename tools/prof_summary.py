"""Summarize a rocprofv3 rocpd results DB into a top-kernels table.

Run ON the GPU box right after rocprofv3 so only the digest (not the
multi-MB trace DB) needs to travel back through gpurun_out:

    rocprofv3 --kernel-trace --stats -d /tmp/prof -- <cmd>
    python tools/prof_summary.py /tmp/prof > gpurun_out/prof_summary.md
"""
import glob
import re
import sqlite3
import sys


def summarize(root: str, limit: int = 40, window_ms: float = 0.0):
    """window_ms > 0: only kernels in the LAST window_ms of the trace —
    the steady-state replays, excluding MIOpen find / warmup."""
    dbs = glob.glob(f"{root}/**/*.db", recursive=True)
    if not dbs:
        raise SystemExit(f"no results db under {root}")
    con = sqlite3.connect(dbs[0])
    tabs = [r[0] for r in con.execute(
        "select name from sqlite_master where type='table'")]
    sfx = re.search(r"rocpd_string_(\w+)", " ".join(tabs)).group(1)
    where = ""
    if window_ms > 0:
        (tmax,) = next(iter(con.execute(
            f'select max(k."end") from rocpd_kernel_dispatch_{sfx} k')))
        where = f'where k.start > {tmax} - {window_ms * 1e6:.0f}'
        print(f"steady-state window: last {window_ms:.0f} ms\n")
    q = f'''
    select ks.display_name, count(*), sum(k."end" - k.start)/1e6,
           avg(k."end" - k.start)/1e3
    from rocpd_kernel_dispatch_{sfx} k
    join rocpd_info_kernel_symbol_{sfx} ks on k.kernel_id = ks.id
    {where}
    group by 1 order by 3 desc
    '''
    rows = list(con.execute(q))
    tot = sum(r[2] for r in rows)
    print(f"| kernel | calls | total_ms | avg_us | % |")
    print(f"|---|---|---|---|---|")
    for name, n, ms, avg in rows[:limit]:
        print(f"| {name[:78]} | {n} | {ms:.2f} | {avg:.1f} "
              f"| {100 * ms / tot:.1f} |")
    print(f"\ntotal kernel time: {tot:.1f} ms over "
          f"{sum(r[1] for r in rows)} dispatches")


if __name__ == "__main__":
    summarize(sys.argv[1],
              int(sys.argv[2]) if len(sys.argv) > 2 else 40,
              float(sys.argv[3]) if len(sys.argv) > 3 else 0.0)
