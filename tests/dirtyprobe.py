"""Dirty-input whole-job probe (GPU box): flights-shaped CSV at ~1% malformed
rows through the PRODUCT engine (Context.csv -> collect) — measures the
parallel host resolver (presolve.py) against VERDICT r1's >=50 M rows/s bar
(round 1 was ~6 M rows/s with the serial resolver at 2% dirty).

Usage: python tests/dirtyprobe.py [mb] [dirty_frac] [resolveProcesses]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tests import extra_data as X  # noqa: E402
import tuplex_amd  # noqa: E402


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    dirty = float(sys.argv[2]) if len(sys.argv) > 2 else 0.01
    procs = int(sys.argv[3]) if len(sys.argv) > 3 else 0

    t0 = time.perf_counter()
    base = X.make_flights_csv(4000, seed=9, bad_frac=dirty)
    hdr_end = base.find(b"\n") + 1
    header, body = base[:hdr_end], base[hdr_end:]
    reps = max(1, (mb << 20) // len(body))
    path = "/tmp/dirtyprobe.csv"
    with open(path, "wb") as f:
        f.write(header)
        for _ in range(reps):
            f.write(body)
    n_rows = 4000 * reps
    print("input %.1f MB, %d rows, %.2f%% dirty, built in %.1fs"
          % (os.path.getsize(path) / 1e6, n_rows, dirty * 100,
             time.perf_counter() - t0), flush=True)

    for use_procs in ([procs] if procs else [1, 0]):
        ctx = tuplex_amd.Context(
            {"tuplex.gpu.resolveProcesses": str(use_procs)})
        ds = (ctx.csv(path)
              .withColumn("delay_ratio", X.fl_ratio)
              .filter(X.fl_carrier)
              .withColumn("code", X.fl_code)
              .selectColumns(["c0", "code", "delay_ratio", "c3"]))
        t0 = time.perf_counter()
        rows = ds.collect()
        dt = time.perf_counter() - t0
        assert ds._last_outcome.mode == "gpu", \
            ds._last_outcome.fallback_reason
        label = "serial" if use_procs == 1 else ("pool(%s)" % (use_procs or
                                                               "auto"))
        print("%s resolver: %.2f s  %.1f M rows/s whole-job  (%d out rows, "
              "exc %s)" % (label, dt, n_rows / dt / 1e6, len(rows),
                           ds._last_outcome.exception_counts), flush=True)


if __name__ == "__main__":
    main()
