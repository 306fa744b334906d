"""ORC/columnar-ingest perf probe (GPU box): the Z1 pipeline fed from
device-resident Arrow columns (no CSV parse, no boundary scan) vs the CSV
path. Usage: python tests/orcprobe.py [rows_millions]"""
import ctypes
import sys
import time

sys.path.insert(0, ".")

import numpy as np  # noqa: E402

from tests.test_codegen_compile import zillow_ops  # noqa: E402
from tests.zillow_data import ZILLOW_COLS, make_zillow_rows  # noqa: E402
from tuplex_amd import codegen, plan  # noqa: E402
from tuplex_amd import ttypes as T  # noqa: E402
from tuplex_amd.engine import GpuLib, TpxResult  # noqa: E402


def main():
    mrows = float(sys.argv[1]) if len(sys.argv) > 1 else 5.0
    base = make_zillow_rows(150000, seed=42, dirty_frac=0.0)
    reps = max(1, int(mrows * 1e6) // len(base))
    rows = base * reps
    n = len(rows)
    print("rows: %.2fM" % (n / 1e6), flush=True)

    glib = GpuLib.get()
    assert glib.device_count() > 0

    # columnar upload: all 10 columns are str except postal (f64-looking str
    # in the csv; here keep ALL as str to match the csv bench's explicit
    # schema str..str + F64 postal)
    col_types = [T.STR, T.STR, T.STR, T.STR, T.F64, T.STR, T.STR, T.STR,
                 T.STR, T.STR]
    sp = plan.build_stage(col_types, list(ZILLOW_COLS), zillow_ops())
    assert sp.compilable, sp.why_not_compilable
    src, desc = codegen.generate_stage(sp, source="col", sink="csv",
                                       csv_info={"null_values": [""]})
    stage = glib.compile_stage(src, desc)

    dev_ptrs = []
    total = 0

    def up(b):
        nonlocal total
        p = glib.lib.tpx_dev_alloc(max(len(b), 1))
        assert p
        buf = (ctypes.c_uint8 * max(len(b), 1)).from_buffer_copy(b or b"\0")
        assert glib.lib.tpx_dev_upload(p, buf, max(len(b), 1)) == 0
        dev_ptrs.append(p)
        total += len(b)
        return p

    t0 = time.time()
    slots = []
    used = sp.used_source_cols
    for k, t in enumerate(col_types):
        if used is not None and k not in used:
            slots += [0, 0, 0]
            continue
        if T.deopt(t) == T.F64:
            v = np.array([float(r[k]) for r in base], dtype=np.float64)
            v = np.tile(v, reps)
            slots += [up(v.tobytes()), 0, 0]
        else:
            cells = [r[k].encode() for r in base]
            blob1 = b"".join(cells)
            lens = np.array([len(c) for c in cells], dtype=np.int64)
            offs1 = np.concatenate([[0], np.cumsum(lens)])
            # tile: offsets shift per repetition
            stride = offs1[-1]
            offs = np.concatenate(
                [offs1[:-1] + i * stride for i in range(reps)] +
                [[stride * reps]])
            slots += [up(offs.astype(np.int64).tobytes()), up(blob1 * reps), 0]
    print("upload %.1f MB in %.1fs" % (total / 1e6, time.time() - t0),
          flush=True)

    for it in range(7):
        t0 = time.perf_counter()
        res = TpxResult()
        arr = (ctypes.c_void_p * len(slots))(*[ctypes.c_void_p(p or 0)
                                               for p in slots])
        rc = glib.lib.tpx_stage_execute_col(stage, arr, len(slots), n, total,
                                            0, 2, ctypes.byref(res))
        assert rc == 0, glib.err()
        wall = (time.perf_counter() - t0) * 1e3
        if it >= 2:
            print("step %d wall %7.2f ms  main %6.3f compact %6.3f write %6.3f"
                  "  -> %.1f M rows/s"
                  % (it, wall, res.t_main_ms, res.t_compact_ms, res.t_write_ms,
                     n / wall / 1e3), flush=True)
        glib.lib.tpx_result_free(ctypes.byref(res))


if __name__ == "__main__":
    main()
