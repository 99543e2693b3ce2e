"""Aggregate a rocprofv3 kernel-trace sqlite .db into a per-kernel
time table (usage: python tools/kernel_stats.py <dir-with-db> [top_n]).
Marks ray_amd's hand-written HIP kernels."""
import glob
import sqlite3
import sys

OURS = ("fa_bwd", "flash_attn", "adamw", "swiglu", "rope_", "rmsnorm",
        "fused_ce", "cross_entropy", "gae_", "vtrace", "nhwc")


def main(path: str, top: int = 30):
    dbs = glob.glob(f"{path}/**/*.db", recursive=True)
    if not dbs:
        print(f"no .db under {path}")
        return 1
    db = sqlite3.connect(dbs[0])
    suf = None
    for (name,) in db.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        if name.startswith("rocpd_kernel_dispatch_"):
            suf = name[len("rocpd_kernel_dispatch_"):]
            break
    if suf is None:
        print("no rocpd_kernel_dispatch_* table")
        return 1
    rows = list(db.execute(f"""
        SELECT s.display_name, COUNT(*) n,
               SUM(d.end-d.start)/1e6 total_ms
        FROM rocpd_kernel_dispatch_{suf} d
        JOIN rocpd_info_kernel_symbol_{suf} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY total_ms DESC LIMIT {int(top)}
    """))
    grand = list(db.execute(f"""
        SELECT SUM(d.end-d.start)/1e6 FROM rocpd_kernel_dispatch_{suf} d
    """))[0][0]
    ours_ms = 0.0
    for name, n, ms in rows:
        tag = ""
        if any(k in name for k in OURS):
            tag = "  <== ray_amd HIP kernel"
            ours_ms += ms
        print(f"{ms:9.1f} ms {ms/grand*100:5.1f}%  n={n:5d}  "
              f"{name[:64]}{tag}")
    print(f"\ntotal GPU kernel time: {grand:.1f} ms; "
          f"ray_amd kernels (top-{top} rows): {ours_ms:.1f} ms "
          f"({ours_ms/grand*100:.1f}%)")
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1],
                  int(sys.argv[2]) if len(sys.argv) > 2 else 30))
