"""Ablation probe (GPU box): how much of tpx_stage_main is CSV parse vs UDF chain.

Runs the Zillow CSV through (A) a passthrough stage (parse + columnar store of all
10 input columns, no UDFs) and (B) the full Z1 pipeline, device-resident, and
prints per-kernel times. Usage: python tests/ablate.py [mb]
"""
import ctypes
import sys

sys.path.insert(0, ".")

from bench import make_input  # noqa: E402
from tests.test_codegen_compile import zillow_ops  # noqa: E402
from tuplex_amd import codegen, csvio, plan  # noqa: E402
from tuplex_amd.engine import GpuLib, TpxResult  # noqa: E402


def run(stage, glib, dev, size, label):
    for i in range(3):
        res = TpxResult()
        rc = glib.lib.tpx_stage_execute_csv_dev(stage, dev, size, 0, 2,
                                                ctypes.byref(res))
        assert rc == 0, glib.err()
        if i == 2:
            print("%-12s main %.3f ms  boundary %.3f  compact %.3f  write %.3f  "
                  "rows %d out %d" % (label, res.t_main_ms, res.t_boundary_ms,
                                      res.t_compact_ms, res.t_write_ms,
                                      res.in_num_rows, res.out_num_rows))
        glib.lib.tpx_result_free(ctypes.byref(res))


def main():
    mb = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
    header, body = make_input(150000, 42, 0.0, mb << 20)
    glib = GpuLib.get()
    assert glib.device_count() > 0
    dev = glib.lib.tpx_dev_alloc(len(body))
    buf = (ctypes.c_uint8 * len(body)).from_buffer_copy(body)
    glib.lib.tpx_dev_upload(dev, buf, len(body))
    del buf

    from tests.zillow_data import ZILLOW_COLS
    from tuplex_amd import ttypes as T
    # EXACTLY the bench stage shape (explicit schema, csv sink)
    col_types = [T.STR, T.STR, T.STR, T.STR, T.F64, T.STR, T.STR, T.STR,
                 T.STR, T.STR]
    names = list(ZILLOW_COLS)

    zo = zillow_ops()
    variants = [
        ("passthrough", [], None),
        ("udf_nofilter", [op for op in zo
                          if op[0] in ("withColumn", "mapColumn")], None),
        ("full", zo, None),
        ("full_noLDS", zo, 16),             # span cap 16B -> all-global path,
                                            # tiny smem -> full occupancy
    ]
    def build(ops):
        sp = plan.build_stage(col_types, names, ops)
        assert sp.compilable, sp.why_not_compilable
        try:
            src, desc = codegen.generate_stage(sp, source="csv", sink="csv",
                                               csv_info={"null_values": [""]})
            sink = "csv"
        except codegen.CodegenError:   # f64 outputs: mem sink for the probe
            src, desc = codegen.generate_stage(sp, source="csv", sink="mem",
                                               csv_info={"null_values": [""]})
            sink = "mem"
        return glib.compile_stage(src, desc), sink

    saved_cap = codegen.StageCodegen.SPAN_CAP
    for label, ops, cap in variants:
        codegen.StageCodegen.SPAN_CAP = cap if cap else saved_cap
        stage, sink = build(ops)
        run(stage, glib, dev, len(body), label + ("[mem]" if sink == "mem" else ""))
    codegen.StageCodegen.SPAN_CAP = saved_cap

    if len(sys.argv) > 2 and sys.argv[2] == "ops":
        # cumulative per-op ablation: prefix of the Z1 chain, csv sink
        prev = None
        for k in range(len(zo) + 1):
            ops = zo[:k]
            # selectColumns only valid at the end; skip bare filter prefixes ok
            stage, sink = build(ops)
            run(stage, glib, dev, len(body),
                ("+%s" % (zo[k - 1][0] + ":" + getattr(zo[k - 1][-1], "__name__",
                                                       str(zo[k - 1][1]))[:16])
                 if k else "none") + ("[mem]" if sink == "mem" else ""))


if __name__ == "__main__":
    main()
