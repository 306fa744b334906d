"""Write-kernel probe (GPU box): time tpx_stage_write variants on the bench
stage (csv sink, device-resident). Usage: python tests/wprobe.py [mb]"""
import ctypes
import sys

sys.path.insert(0, ".")

from bench import make_input, build_stage  # noqa: E402
from tuplex_amd import codegen  # noqa: E402
from tuplex_amd.engine import GpuLib, TpxResult  # noqa: E402


def run(stage, glib, dev, size, label):
    ts = []
    for i in range(4):
        res = TpxResult()
        rc = glib.lib.tpx_stage_execute_csv_dev(stage, dev, size, 0, 2,
                                                ctypes.byref(res))
        assert rc == 0, glib.err()
        if i:
            ts.append((res.t_write_ms, res.t_main_ms, res.t_compact_ms))
        glib.lib.tpx_result_free(ctypes.byref(res))
    w = min(t[0] for t in ts)
    m = min(t[1] for t in ts)
    c = min(t[2] for t in ts)
    print("%-14s write %.3f ms  main %.3f  compact %.3f" % (label, w, m, c))


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
    header, body = make_input(150000, 42, 0.0, mb << 20)
    glib = GpuLib.get()
    assert glib.device_count() > 0
    dev = glib.lib.tpx_dev_alloc(len(body))
    buf = (ctypes.c_uint8 * len(body)).from_buffer_copy(body)
    glib.lib.tpx_dev_upload(dev, buf, len(body))
    del buf

    saved = codegen.StageCodegen.WRITE_CAP
    for cap in [8192, 4096, 2048, 16]:
        codegen.StageCodegen.WRITE_CAP = cap
        src, desc = build_stage()
        stage = glib.compile_stage(src, desc)
        run(stage, glib, dev, len(body), "wcap=%d" % cap)
    codegen.StageCodegen.WRITE_CAP = saved


if __name__ == "__main__":
    main()
