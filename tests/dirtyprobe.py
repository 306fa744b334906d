"""Dirty-input whole-job probe (GPU box): Zillow-shaped CSV with malformed
rows through the PRODUCT engine (Context.csv -> tocsv; output formats on
device, so the job measures GPU + the host resolver, not python row
materialization — the reference's benchmarks use file sinks for the same
reason). Round 1: ~6 M rows/s whole-job at 2% dirty with the serial CPython
resolver; VERDICT r1 asks >=50 M rows/s at ~1%.

Usage: python tests/dirtyprobe.py [mb] [dirty_frac] [resolveProcesses]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bench import make_input  # noqa: E402
from tests.test_codegen_compile import zillow_ops  # noqa: E402
from tests.pipelines import apply_ops  # noqa: E402
import tuplex_amd  # noqa: E402


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
    dirty = float(sys.argv[2]) if len(sys.argv) > 2 else 0.02
    procs = int(sys.argv[3]) if len(sys.argv) > 3 else 0

    t0 = time.perf_counter()
    header, body = make_input(150000, 42, dirty, mb << 20)
    path = "/tmp/dirtyprobe.csv"
    with open(path, "wb") as f:
        f.write(header)
        f.write(body)
    n_rows = body.count(b"\n")
    print("input %.1f MB, %d rows, %.2f%% dirty, built in %.1fs"
          % (os.path.getsize(path) / 1e6, n_rows, dirty * 100,
             time.perf_counter() - t0), flush=True)

    for use_procs in ([procs] if procs else [1, 0]):
        label = "serial" if use_procs == 1 else ("pool(%s)" % (use_procs or
                                                               "auto"))
        for it in range(2):  # 2nd iteration = warm persistent pool
            ctx = tuplex_amd.Context(
                {"tuplex.gpu.resolveProcesses": str(use_procs)})
            ds = apply_ops(ctx.csv(path), zillow_ops())
            t0 = time.perf_counter()
            ds.tocsv("/tmp/dirtyprobe_out.csv")
            dt = time.perf_counter() - t0
            assert ds._last_outcome.mode == "gpu", \
                ds._last_outcome.fallback_reason
            exc = ds._last_outcome.exception_counts
            print("%s resolver (pass %d): %.2f s  %.1f M rows/s whole-job"
                  "  (exc %s)"
                  % (label, it, dt, n_rows / dt / 1e6,
                     dict(list(exc.items())[:3])), flush=True)
    os.unlink(path)


if __name__ == "__main__":
    main()
