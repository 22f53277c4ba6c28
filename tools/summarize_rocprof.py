#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into a small markdown table (run on
the GPU box; the raw db stays there — only the summary is copied back)."""

import glob
import sys


def main(db_glob, out_path, note=""):
    import sqlite3
    paths = sorted(glob.glob(db_glob))
    assert paths, f"no db matches {db_glob}"
    c = sqlite3.connect(paths[-1])
    rows = list(c.execute(
        """select ks.display_name, count(*), sum(kd.end-kd.start)/1e6,
                  avg(kd.end-kd.start)/1e3
           from rocpd_kernel_dispatch kd
           join rocpd_info_kernel_symbol ks on kd.kernel_id = ks.id
           group by ks.display_name order by 3 desc limit 40"""))
    total, n_disp = list(c.execute(
        "select sum(end-start)/1e6, count(*) from rocpd_kernel_dispatch"))[0]
    with open(out_path, "w") as f:
        f.write(f"# rocprofv3 kernel stats\n\n{note}\n\n")
        f.write(f"Total kernel time {total:.1f} ms / {n_disp} dispatches.\n\n")
        f.write("| kernel | calls | total ms | avg us | % |\n|---|---|---|---|---|\n")
        for k, n, tot, avg in rows:
            f.write(f"| `{k[:100]}` | {n} | {tot:.2f} | {avg:.1f} "
                    f"| {100 * tot / total:.1f} |\n")
    print(f"wrote {out_path}: {total:.1f} ms, {n_disp} dispatches")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], sys.argv[3] if len(sys.argv) > 3 else "")
