#!/usr/bin/env python3
"""Kernel composition of the trailing decode window of a rocprofv3 rocpd
trace (bench.py runs end with the timed decode loop, so the last WINDOW
seconds of device time are pure decode steps).

Usage: python scripts/decode_window.py <results.db|dir> [window_s]
"""

import glob
import sqlite3
import sys


def main() -> None:
    target = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out"
    win = float(sys.argv[2]) if len(sys.argv) > 2 else 0.65
    files = ([target] if target.endswith(".db")
             else sorted(glob.glob(f"{target}/**/*results.db",
                                   recursive=True)))
    if not files:
        print(f"no results.db under {target}")
        return
    db = sqlite3.connect(files[-1])
    suf = [r[0].replace("rocpd_kernel_dispatch", "") for r in db.execute(
        "select name from sqlite_master where type='table'")
        if r[0].startswith("rocpd_kernel_dispatch")][0]
    tmax = list(db.execute(
        f"select max(end) from rocpd_kernel_dispatch{suf}"))[0][0]
    w0 = tmax - int(win * 1e9)
    q = f"""
    select ks.display_name, count(*) as calls, sum(k.end-k.start)/1e6 as ms
    from rocpd_kernel_dispatch{suf} k
    join rocpd_info_kernel_symbol{suf} ks on k.kernel_id=ks.id
    where k.start > {w0}
    group by ks.display_name order by ms desc limit 14
    """
    tot = list(db.execute(
        f"select sum(end-start)/1e6 from rocpd_kernel_dispatch{suf} "
        f"where start > {w0}"))[0][0]
    print(f"{files[-1]}: decode window {win}s wall, {tot:.0f} ms device")
    for name, calls, ms in db.execute(q):
        print(f"{ms:8.1f} ms {100*ms/tot:5.1f}% x{calls:>5}  {name[:82]}")


if __name__ == "__main__":
    main()
