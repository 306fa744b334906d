"""Larger-than-resident streaming probe (GPU box): run the Zillow pipeline
over an input bigger than tuplex.gpu.residentMaxSize so the engine takes the
ranged path (256 MB GPU ranges: per-range H2D -> kernels -> D2H, HBM
footprint bounded by one range's working set — the device-side analog of the
reference's 128 KB read-buffer streaming, CSVReader.cc:390).

Usage: python tests/streamprobe.py [gb] [resident_max]
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
    gb = float(sys.argv[1]) if len(sys.argv) > 1 else 8.0
    resident = sys.argv[2] if len(sys.argv) > 2 else "1GB"
    path = os.environ.get("TPX_STREAM_FILE", "/tmp/stream_in.csv")
    outp = "/tmp/stream_out.csv"

    t0 = time.perf_counter()
    header, body = make_input(150000, 42, 0.0, 1 << 30)
    want = int(gb * (1 << 30))
    with open(path, "wb") as f:
        f.write(header)
        written = 0
        while written < want:
            f.write(body)
            written += len(body)
    rows_in = body.count(b"\n") * (written // len(body))
    print("input %.1f GB (%d rows) built in %.1fs"
          % (written / 2**30, rows_in, time.perf_counter() - t0), flush=True)

    ctx = tuplex_amd.Context({"tuplex.gpu.residentMaxSize": resident})
    ds = apply_ops(ctx.csv(path), zillow_ops())
    t0 = time.perf_counter()
    ds.tocsv(outp)
    dt = time.perf_counter() - t0
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    m = ds._last_outcome.metrics
    print("ranged run: %.2f s  %.1f M rows/s e2e  chunks=%s  kernel %.0f ms  "
          "h2d %.0f ms  out %.1f MB"
          % (dt, rows_in / dt / 1e6, m.get("chunks"), m.get("t_kernel_ms", 0),
             m.get("t_h2d_ms", 0), os.path.getsize(outp) / 1e6), flush=True)
    os.unlink(path)
    os.unlink(outp)


if __name__ == "__main__":
    main()
