"""Summarize a rocprofv3 rocpd SQLite database into a markdown kernel-time
table (committed under profiles/ as benchmark evidence).

Usage: python benchmarks/dump_profile.py <results.db> <out.md> [steps]
"""

import sqlite3
import sys


def main():
    dbf, out = sys.argv[1], sys.argv[2]
    steps = int(sys.argv[3]) if len(sys.argv) > 3 else None
    db = sqlite3.connect(dbf)
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE "
        "'rocpd_kernel_dispatch%'")][0]
    sfx = t[len("rocpd_kernel_dispatch"):]
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(d.end - d.start)/1e6,
           AVG(d.end-d.start)/1e3
    FROM rocpd_kernel_dispatch{sfx} d
    JOIN rocpd_info_kernel_symbol{sfx} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC
    """
    rows = list(cur.execute(q))
    tot = sum(r[2] for r in rows)
    with open(out, "w") as f:
        f.write(f"# Kernel-time summary: {dbf}\n\n")
        f.write(f"Total GPU kernel time: {tot:.1f} ms")
        if steps:
            f.write(f" over {steps} steps = {tot/steps:.2f} ms/step")
        f.write("\n\n| kernel | calls | total ms | avg us | % |\n")
        f.write("|---|---|---|---|---|\n")
        for name, calls, ms, avg in rows[:30]:
            # demangled form is "void (anonymous namespace)::name<T>(args)";
            # drop the namespace wrapper before cutting at the arg list so
            # the kernel's own name survives
            n = name.replace("(anonymous namespace)::", "")
            short = n.split("(")[0][:100] if "(" not in n[:6] else n[:100]
            f.write(f"| `{short}` | {calls} | {ms:.2f} | {avg:.1f} | "
                    f"{100*ms/tot:.1f} |\n")
    print(f"wrote {out}: {tot:.1f} ms total")


if __name__ == "__main__":
    main()
