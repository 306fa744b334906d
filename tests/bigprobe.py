"""Per-step timing probe at large scale (GPU box): where does the time go as
input size grows? Usage: python tests/bigprobe.py [mb]"""
import ctypes
import sys
import time

sys.path.insert(0, ".")

from bench import build_stage, make_input  # noqa: E402
from tuplex_amd.engine import GpuLib, TpxResult  # noqa: E402


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    header, body = make_input(150000, 42, 0.0, mb << 20)
    glib = GpuLib.get()
    assert glib.device_count() > 0
    src, desc = build_stage()
    stage = glib.compile_stage(src, desc)
    dev = glib.lib.tpx_dev_alloc(len(body))
    assert dev
    buf = (ctypes.c_uint8 * len(body)).from_buffer_copy(body)
    glib.lib.tpx_dev_upload(dev, buf, len(body))
    del buf

    for it in range(7):
        t0 = time.perf_counter()
        res = TpxResult()
        rc = glib.lib.tpx_stage_execute_csv_dev(stage, dev, len(body), 0, 2,
                                                ctypes.byref(res))
        assert rc == 0, glib.err()
        wall = (time.perf_counter() - t0) * 1e3
        print("step %d wall %8.2f ms  boundary %7.3f main %7.3f compact %7.3f "
              "write %7.3f  rows %d out %d"
              % (it, wall, res.t_boundary_ms, res.t_main_ms, res.t_compact_ms,
                 res.t_write_ms, res.in_num_rows, res.out_num_rows),
              flush=True)
        glib.lib.tpx_result_free(ctypes.byref(res))


if __name__ == "__main__":
    main()
