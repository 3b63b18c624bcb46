"""Extract a kernel-stats summary (per-kernel time, calls, own/lib
classification, own-kernel share) from a rocprofv3 results DB into a
committed text file under profiles/."""
import glob
import sqlite3
import sys


def summarize(db_glob: str, out_path: str, steps: int, title: str) -> None:
    db_path = sorted(glob.glob(db_glob))[-1]
    db = sqlite3.connect(db_path)
    sfx = [
        r[0]
        for r in db.execute(
            "SELECT name FROM sqlite_master WHERE type='table' "
            "AND name LIKE 'rocpd_kernel_dispatch%'"
        )
    ][0].replace("rocpd_kernel_dispatch_", "")
    rows = list(
        db.execute(
            f"SELECT s.kernel_name, s.display_name, COUNT(*), "
            f"SUM(d.end-d.start)/1e6 "
            f"FROM rocpd_kernel_dispatch_{sfx} d "
            f"JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id=s.id "
            f"GROUP BY s.kernel_name ORDER BY 4 DESC"
        )
    )
    tot = sum(r[3] for r in rows)
    own = 0.0
    lines = [
        title,
        f"total kernel time: {tot:.2f} ms over {steps} steps "
        f"({tot/steps:.3f} ms/step)",
        "",
        f"{'ms':>8} {'ms/step':>8} {'calls':>6} {'tag':>4}  kernel",
    ]
    for kn, dn, c, ms in rows:
        is_own = kn.startswith("_Z") and not kn.startswith(
            ("_ZN2at", "_ZN7rocprim", "_ZN12_GLOBAL")
        )
        if is_own:
            own += ms
        if ms > tot * 0.002:
            name = (dn if dn and dn != "void" else kn)[:95]
            lines.append(
                f"{ms:8.2f} {ms/steps:8.3f} {c:6d} "
                f"{'OWN' if is_own else 'LIB':>4}  {name}"
            )
    lines += [
        "",
        f"OWN-kernel share of GPU time: {100*own/tot:.1f}% "
        f"(own {own:.2f} ms / lib {tot-own:.2f} ms)",
    ]
    with open(out_path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print("\n".join(lines[:6]))
    print(lines[-1])


if __name__ == "__main__":
    summarize(sys.argv[1], sys.argv[2], int(sys.argv[3]), sys.argv[4])
