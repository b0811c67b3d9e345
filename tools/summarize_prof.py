#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db: top kernels, memcpy totals, and the
SDMA-copy/compute overlap fraction per process (evidence that the chunk
pipeline overlaps wire traffic with reduction kernels)."""
import glob
import sqlite3
import sys


def union_busy(intervals):
    if not intervals:
        return 0, []
    intervals.sort()
    merged = [list(intervals[0])]
    for s, e in intervals[1:]:
        if s <= merged[-1][1]:
            merged[-1][1] = max(merged[-1][1], e)
        else:
            merged.append([s, e])
    return sum(e - s for s, e in merged), merged


def overlap(a_merged, b_merged):
    i = j = tot = 0
    while i < len(a_merged) and j < len(b_merged):
        s = max(a_merged[i][0], b_merged[j][0])
        e = min(a_merged[i][1], b_merged[j][1])
        if s < e:
            tot += e - s
        if a_merged[i][1] < b_merged[j][1]:
            i += 1
        else:
            j += 1
    return tot


def main(path):
    dbs = sorted(glob.glob(path + "/**/*_results.db", recursive=True))
    if not dbs:
        print("no results.db under", path)
        return
    for db in dbs:
        c = sqlite3.connect(db)
        tabs = [r[0] for r in c.execute(
            "select name from sqlite_master where type='table'")]
        sfx = None
        for t in tabs:
            if t.startswith("rocpd_kernel_dispatch_"):
                sfx = t[len("rocpd_kernel_dispatch_"):]
        if sfx is None:
            continue
        print(f"== {db}")
        print("-- top kernels by total time")
        for row in c.execute(
                f"select ks.display_name, count(*), "
                f"sum(kd.end-kd.start)/1e6, avg(kd.end-kd.start)/1e3 "
                f"from rocpd_kernel_dispatch_{sfx} kd "
                f"join rocpd_info_kernel_symbol_{sfx} ks "
                f"on kd.kernel_id = ks.id "
                f"group by ks.display_name order by 3 desc limit 10"):
            print(f"  {row[0][:64]:64s} n={row[1]:5d} "
                  f"tot_ms={row[2]:9.2f} avg_us={row[3]:8.1f}")
        # Same-device "peer" copies on a timeshared box run as blit
        # KERNELS (__amd_rocclr_copyBuffer), not SDMA memcpies: compute
        # copy-kernel vs reduce-kernel overlap per pid too.
        kinds = {}
        for kid, name in c.execute(
                f"select id, display_name from "
                f"rocpd_info_kernel_symbol_{sfx}"):
            if "copyBuffer" in name:
                kinds[kid] = "copy"
            elif "reduce" in name.lower():
                kinds[kid] = "reduce"
        krows = list(c.execute(
            f"select pid, kernel_id, start, end from "
            f"rocpd_kernel_dispatch_{sfx}"))
        for pid in sorted(set(r[0] for r in krows)):
            cints = [(s, e) for p, k, s, e in krows
                     if p == pid and kinds.get(k) == "copy"]
            rints = [(s, e) for p, k, s, e in krows
                     if p == pid and kinds.get(k) == "reduce"]
            if not cints or not rints:
                continue
            cb, cm = union_busy(cints)
            rb, rm = union_busy(rints)
            ov = overlap(cm, rm)
            print(f"  pid {pid}: blit-copy busy={cb/1e6:.2f}ms "
                  f"reduce busy={rb/1e6:.2f}ms "
                  f"copy∩reduce={ov/1e6:.2f}ms "
                  f"({100*ov/max(min(cb,rb),1):.0f}% of the shorter)")
        try:
            rows = list(c.execute(
                f"select pid, start, end, size from rocpd_memory_copy_{sfx}"))
        except sqlite3.OperationalError:
            rows = []
        pids = sorted(set(r[0] for r in rows))
        print("-- memcpy + overlap per pid")
        for pid in pids:
            copies = [(r[1], r[2]) for r in rows if r[0] == pid]
            nbytes = sum(r[3] for r in rows if r[0] == pid)
            cb, cm = union_busy(copies)
            kerns = [(r[0], r[1]) for r in c.execute(
                f"select start, end from rocpd_kernel_dispatch_{sfx} "
                f"where pid = ?", (pid,))]
            kb, km = union_busy(kerns)
            ov = overlap(cm, km)
            span = 0
            allints = copies + kerns
            if allints:
                span = max(e for _, e in allints) - min(
                    s for s, _ in allints)
            print(f"  pid {pid}: copies n={len(copies)} "
                  f"bytes={nbytes/1e6:.1f}MB busy={cb/1e6:.2f}ms "
                  f"({nbytes/max(cb,1)*1e9/1e9:.1f}GB/s) | kernels "
                  f"busy={kb/1e6:.2f}ms | copy∩kernel={ov/1e6:.2f}ms "
                  f"({100*ov/max(cb,1):.0f}% of copy time) | "
                  f"span={span/1e6:.1f}ms")


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else "gpurun_out")
