#!/usr/bin/env python3
"""bench.py — headline benchmark: Zillow map/filter TransformStage on MI355X.

BASELINE.json metric: rows/s (+ achieved HBM GB/s) on the Zillow pipeline
(benchmarks/zillow/Z1/runtuplex.py:192-205 operator chain), synthetic
Zillow-shaped CSV, csv sink. One "step" = one pass of the fused
GPU pipeline (row-boundary scan -> fused parse+UDF -> compaction -> CSV write)
over the rank's resident input buffer; inputs are in HBM when the timed region
starts and outputs stay device-resident (the device analog of the reference's
memory partitions; PCIe-inclusive rates are reported in DESIGN.md §Measurement).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
         [--mb-per-gpu MB] [--base-rows R] [--dirty F] [--cpu-seconds S]
         [--pmc-summary FILE]
Launched for N>1 by torchrun (one rank per GPU; RANK/LOCAL_RANK/WORLD_SIZE env).
Rank 0 prints ONE JSON line.
"""
import argparse
import ctypes
import json
import os
import subprocess
import sys
import time

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, HERE)

HBM_PEAK = 8.0e12  # MI355X spec peak (MI355X_MICROARCH.md); 6.3e12 achievable


def log(*a):
    print(*a, file=sys.stderr, flush=True)


def make_input(base_rows, seed, dirty, target_bytes):
    from tests.zillow_data import make_zillow_csv_bytes
    data, _ = make_zillow_csv_bytes(base_rows, seed=seed, dirty_frac=dirty,
                                    header=True)
    p = data.find(b"\n")
    header, body = data[:p + 1], data[p + 1:]
    reps = max(1, target_bytes // len(body))
    return header, body * reps


def build_stage():
    from tests.test_codegen_compile import zillow_ops
    from tuplex_amd import codegen, plan
    from tuplex_amd import ttypes as T
    sp = plan.build_stage(
        [T.STR, T.STR, T.STR, T.STR, T.F64, T.STR, T.STR, T.STR, T.STR, T.STR],
        ["title", "address", "city", "state", "postal_code", "price",
         "facts and features", "real estate provider", "url", "sales_date"],
        zillow_ops())
    assert sp.compilable, sp.why_not_compilable
    src, desc = codegen.generate_stage(sp, source="csv", sink="csv",
                                       csv_info={"null_values": [""]})
    return src, desc


def cpu_baseline_leg(body, seconds):
    """Time the oracle C port (kind 'port') on the host cores of this box —
    bounded sample. Primary number = ALL host cores (the reference runs its
    compiled path at executorCount = hw threads, LocalBackend.cc:46); a
    single-core leg is reported alongside as value_1core."""
    binp = os.path.join(HERE, "oracle", "czillow")
    if not os.path.exists(binp):
        try:
            subprocess.check_call(["make", "-C", os.path.join(HERE, "oracle"),
                                   "-s"])
        except Exception as e:  # noqa: BLE001
            log("cpu_baseline build failed:", e)
            return None
    sample = body[:64 << 20]
    nl = sample.rfind(b"\n")
    sample = sample[:nl + 1]
    tmp = "/tmp/tpx_bench_sample.csv"
    with open(tmp, "wb") as f:
        f.write(b"h1,h2,h3,h4,h5,h6,h7,h8,h9,h10\n")  # czillow skips line 1
        f.write(sample)
    ncores = os.cpu_count() or 1
    try:
        out = subprocess.check_output([binp, tmp, str(seconds), str(ncores)],
                                      timeout=seconds * 4 + 60)
        r = json.loads(out)
        out1 = subprocess.check_output([binp, tmp, str(max(seconds / 2, 2)),
                                        "1"], timeout=seconds * 4 + 60)
        r1 = json.loads(out1)
        return {"value": r["rows_per_s"], "unit": "rows/s", "cores": ncores,
                "kind": "port",
                "value_1core": r1["rows_per_s"],
                "sample": "%d MB of the same synthetic Zillow CSV; %d s on all "
                          "%d host cores (pthread range shards), plus a "
                          "single-core leg" % (len(sample) >> 20, seconds,
                                               ncores)}
    except Exception as e:  # noqa: BLE001
        log("cpu_baseline failed:", e)
        return None


def e2e_leg(header, body, mb, local_rank=0, dist=None, rank=0, world=1):
    """End-to-end file->file tocsv: disk read + H2D + kernels + D2H + disk
    write ALL inside the timed region (SURVEY.md §8d's IO-included number),
    through the product engine (Context.csv -> tocsv), reported beside the
    resident-kernel headline in the same JSON line. At world>1 every rank runs
    it together on ONE shared input file (total world*mb MB): the engine
    itself shards chunks across ranks (csvio.run_csv dist path) and each rank
    writes its own part file — this is the product multi-GPU path, measured."""
    import shutil
    import tuplex_amd
    from tests.test_codegen_compile import zillow_ops
    from tests.pipelines import apply_ops
    tdir = "/tmp/tpx_e2e"
    os.makedirs(tdir, exist_ok=True)
    inp = os.path.join(tdir, "in.csv")
    outp = os.path.join(tdir, "out.csv" if world == 1 else "out_parts")
    want = mb << 20
    try:
        if rank == 0:
            with open(inp, "wb") as f:
                f.write(header)
                written = 0
                while written < want * world:
                    f.write(body)
                    written += len(body)
        if dist:
            dist.barrier()
        t0 = time.perf_counter()
        ok = 1
        try:
            ctx = tuplex_amd.Context({"tuplex.gpu.device": str(local_rank)})
            ds = apply_ops(ctx.csv(inp), zillow_ops())
            ds.tocsv(outp)
            if ds._last_outcome.mode != "gpu":
                log("e2e leg fell back:", ds._last_outcome.fallback_reason)
                ok = 0
        except Exception as e:  # noqa: BLE001 — stay collective on failure
            log("e2e leg failed:", e)
            ok = 0
        t1 = time.perf_counter()
        wall = t1 - t0
        if dist:
            # agree on success + MAX wall collectively so no rank bails out
            # of a pending collective (deadlock) when another rank failed
            import torch
            red_dev = ("cuda" if dist.get_backend() == "nccl" else "cpu")
            t = torch.tensor([float(ok), wall], device=red_dev)
            dist.all_reduce(t[:1], op=dist.ReduceOp.MIN)
            dist.all_reduce(t[1:], op=dist.ReduceOp.MAX)
            ok = int(t[0].item())
            wall = float(t[1].item())
        if not ok:
            return None
        if rank != 0:
            return None
        with open(inp, "rb") as f:
            f.seek(len(header))
            rows = sum(chunk.count(b"\n")
                       for chunk in iter(lambda: f.read(1 << 24), b""))
        sz = os.path.getsize(inp)
        return {"e2e_rows_per_s": rows / wall,
                "e2e_mb": sz >> 20,
                "e2e_gb_per_s": sz / wall / 1e9,
                "e2e_seconds": wall, "e2e_engine_ranks": world}
    except Exception as e:  # noqa: BLE001
        log("e2e leg failed:", e)
        return None
    finally:
        if rank == 0:
            shutil.rmtree(tdir, ignore_errors=True)


def parse_pmc_dbs(paths):
    """HBM traffic per tpx_stage_main launch from rocprofv3 rocpd sqlite dbs
    (separate FETCH_SIZE and WRITE_SIZE passes — they don't fit one TCC pass).
    gfx950: FETCH_SIZE reports half the bytes of wide coalesced reads
    (MI355X_MICROARCH.md §HBM) -> traffic = (2*FETCH + WRITE) * 1024 per launch.
    Valid only when the PMC runs used the same workload config as this bench."""
    import glob as _glob
    import sqlite3
    counters = {}
    try:
        for p in paths:
            for dbp in _glob.glob(os.path.join(p, "**", "*_results.db"),
                                  recursive=True) or [p]:
                db = sqlite3.connect(dbp)
                cur = db.cursor()
                tabs = [r[0] for r in cur.execute(
                    "SELECT name FROM sqlite_master WHERE type='table'")]
                kd = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")]
                if not kd:
                    continue
                u = kd[0].replace("rocpd_kernel_dispatch_", "")
                q = ("SELECT pi.name, AVG(pe.value) "
                     "FROM rocpd_pmc_event_%s pe "
                     "JOIN rocpd_kernel_dispatch_%s k ON pe.event_id=k.event_id "
                     "JOIN rocpd_info_kernel_symbol_%s ks ON k.kernel_id=ks.id "
                     "JOIN rocpd_info_pmc_%s pi ON pe.pmc_id=pi.id "
                     "WHERE ks.kernel_name LIKE 'tpx_stage_main%%' "
                     "GROUP BY pi.name" % (u, u, u, u))
                for name, avg in cur.execute(q):
                    counters[name] = avg
        if "FETCH_SIZE" not in counters and "WRITE_SIZE" not in counters:
            return None
        return (2.0 * counters.get("FETCH_SIZE", 0.0)
                + counters.get("WRITE_SIZE", 0.0)) * 1024.0
    except Exception as e:  # noqa: BLE001
        log("pmc db parse failed:", e)
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--mb-per-gpu", type=int, default=1024)
    ap.add_argument("--base-rows", type=int, default=150000)
    ap.add_argument("--dirty", type=float, default=0.0)
    ap.add_argument("--cpu-seconds", type=float, default=10.0)
    ap.add_argument("--pmc-db", type=str, nargs="*", default=None,
                    help="rocprofv3 output dirs (FETCH_SIZE / WRITE_SIZE passes "
                         "at THIS workload config) for roofline.traffic")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--no-e2e", action="store_true")
    ap.add_argument("--e2e-mb", type=int, default=None,
                    help="end-to-end file->file leg size (default: mb-per-gpu)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = world if world > 1 else args.gpus

    dist = None
    torch = None
    backend = os.environ.get("TPX_BENCH_BACKEND", "nccl")
    if world > 1:
        import torch  # noqa: F811
        import torch.distributed as dist  # noqa: F811
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
        dist.init_process_group(backend)

    from tuplex_amd.engine import GpuLib, TpxResult
    glib = GpuLib.get()
    ndev = glib.device_count()
    if ndev == 0:
        log("FATAL: no HIP device; bench requires an MI355X")
        sys.exit(2)
    glib.lib.tpx_set_device(local_rank % ndev)

    t0 = time.time()
    header, body = make_input(args.base_rows, seed=42 + rank, dirty=args.dirty,
                              target_bytes=args.mb_per_gpu << 20)
    log("rank %d: input %.1f MB generated in %.1fs"
        % (rank, len(body) / 1e6, time.time() - t0))

    src, desc = build_stage()
    t0 = time.time()
    stage = glib.compile_stage(src, desc)
    log("rank %d: stage compiled in %.1fs" % (rank, time.time() - t0))

    dev = glib.lib.tpx_dev_alloc(len(body))
    assert dev, "device alloc failed"
    buf = (ctypes.c_uint8 * len(body)).from_buffer_copy(body)
    t0 = time.time()
    rc = glib.lib.tpx_dev_upload(dev, buf, len(body))
    assert rc == 0, glib.err()
    t_upload = time.time() - t0
    del buf

    def step():
        res = TpxResult()
        rc = glib.lib.tpx_stage_execute_csv_dev(stage, dev, len(body), 0, 2,
                                                ctypes.byref(res))
        if rc != 0:
            raise RuntimeError(glib.err())
        out = {
            "in_rows": res.in_num_rows, "out_rows": res.out_num_rows,
            "bytes_in": res.bytes_in, "bytes_out": res.bytes_out,
            "t_kernel_ms": res.t_kernel_ms, "t_boundary_ms": res.t_boundary_ms,
            "t_main_ms": res.t_main_ms, "t_compact_ms": res.t_compact_ms,
            "t_write_ms": res.t_write_ms, "exc": res.exc_num_rows,
        }
        glib.lib.tpx_result_free(ctypes.byref(res))
        return out

    for _ in range(args.warmup):
        last = step()

    if dist:
        dist.barrier()
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    kstats = []
    for _ in range(args.steps):
        kstats.append(step())
    if dist:
        torch.cuda.synchronize()
        dist.barrier()
    t1 = time.perf_counter()
    wall = t1 - t0
    if dist:
        red_dev = "cuda" if backend == "nccl" else "cpu"
        tt = torch.tensor([wall], device=red_dev)
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        wall = float(tt.item())
        rows_t = torch.tensor([float(sum(k["in_rows"] for k in kstats))],
                              device=red_dev)
        dist.all_reduce(rows_t, op=dist.ReduceOp.SUM)
        total_rows = float(rows_t.item())
        bytes_t = torch.tensor([float(sum(k["bytes_in"] + k["bytes_out"]
                                          for k in kstats))], device=red_dev)
        dist.all_reduce(bytes_t, op=dist.ReduceOp.SUM)
        total_bytes = float(bytes_t.item())
    else:
        total_rows = float(sum(k["in_rows"] for k in kstats))
        total_bytes = float(sum(k["bytes_in"] + k["bytes_out"] for k in kstats))

    # e2e leg is collective at world>1 (engine-path sharding) — every rank
    # must enter it before non-zero ranks exit
    e2e = None
    if not args.no_e2e:
        e2e = e2e_leg(header, body, args.e2e_mb or args.mb_per_gpu,
                      local_rank % ndev, dist=dist, rank=rank, world=world)
    if rank != 0:
        return

    last = kstats[-1]
    value = total_rows / wall
    # roofline: whole-path kernel time (inputs+outputs resident in HBM);
    # algorithmic bytes = csv bytes in + csv text bytes out, once each
    t_kernel_s = sum(k["t_kernel_ms"] + k["t_boundary_ms"] for k in kstats) / 1e3
    algo_bytes = last["bytes_in"] + last["bytes_out"]
    achieved = (sum(k["bytes_in"] + k["bytes_out"] for k in kstats) / t_kernel_s
                if t_kernel_s > 0 else 0.0)
    traffic = parse_pmc_dbs(args.pmc_db) if args.pmc_db else None
    roofline = {
        "bound": "hbm",
        "achieved": achieved / 1e9,
        "peak": HBM_PEAK / 1e9,
        "unit": "GB/s",
        "frac": achieved / HBM_PEAK,
        "traffic": traffic,  # HBM bytes per tpx_stage_main launch (PMC), or null
        "dominant_kernel": "tpx_stage_main",
        "per_step_ms": {"boundary": last["t_boundary_ms"],
                        "main": last["t_main_ms"],
                        "compact": last["t_compact_ms"],
                        "write": last["t_write_ms"]},
    }
    cpu = None
    if not args.no_cpu_baseline and rank == 0 and world <= 1:
        cpu = cpu_baseline_leg(body, args.cpu_seconds)
    line = {
        "metric": "rows/s",
        "value": value,
        "unit": "rows/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": wall * 1e3 / args.steps,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "u8",
        "data": "synthetic",
        "config": {
            "workload": "zillow-Z1 csv->map/withColumn/filter->tocsv "
                        "(BASELINE.json configs[1])",
            "mb_per_gpu": args.mb_per_gpu,
            "rows_per_step_per_gpu": last["in_rows"],
            "bytes_per_row_in": last["bytes_in"] / max(last["in_rows"], 1),
            "bytes_per_row_out": last["bytes_out"] / max(last["in_rows"], 1),
            "selectivity": last["out_rows"] / max(last["in_rows"], 1),
            "dirty_frac": args.dirty,
            "sink": "csv-device-resident",
            "upload_s": t_upload,
            **(e2e or {}),
        },
        "roofline": roofline,
        "cpu_baseline": cpu,
        "gb_per_s": achieved / 1e9,
    }
    print(json.dumps(line), flush=True)


if __name__ == "__main__":
    main()
