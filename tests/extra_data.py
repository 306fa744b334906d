"""Synthetic data for the non-Zillow BASELINE configs: TPC-H Q6 lineitem
(configs[3]), flights-shaped wide CSV (configs[2]), weblog lines (configs[4])."""
import random

LINEITEM_COLS = ["l_quantity", "l_extendedprice", "l_discount", "l_shipdate"]


def make_lineitem_csv(n, seed=42):
    """4-col preprocessed lineitem (benchmarks/tpch/Q06/runtuplex.py:77-86 form),
    '|'-delimited, no header."""
    rng = random.Random(seed)
    out = []
    for _ in range(n):
        qty = rng.randint(1, 50)
        price = round(rng.uniform(900.0, 105000.0), 2)
        disc = round(rng.uniform(0.0, 0.1), 2)
        date = rng.randint(19920101, 19981231)
        out.append("%d|%.2f|%.2f|%d" % (qty, price, disc, date))
    return ("\n".join(out) + "\n").encode()


def q6_filter_date(x):
    return 19940101 <= x["l_shipdate"] < 19940101 + 10000


def q6_filter_disc(x):
    return 0.06 - 0.01 <= x["l_discount"] <= 0.06 + 0.01


def q6_filter_qty(x):
    return x["l_quantity"] < 24


def q6_comb(a, b):
    return a + b


def q6_agg(a, x):
    return a + x["l_extendedprice"] * x["l_discount"]


def q6_ops():
    """benchmarks/tpch/Q06/runtuplex.py:81-85 operator chain."""
    return [
        ("filter", q6_filter_date),
        ("filter", q6_filter_disc),
        ("filter", q6_filter_qty),
        ("aggregate", q6_comb, q6_agg, 0.0),
    ]


FLIGHT_NCOLS = 110


def make_flights_csv(n, seed=7, bad_frac=0.01):
    """110-column mixed int/float/str rows (~540 B/row like the reference's
    flights sample), ~1% malformed rows exercising the resolver."""
    rng = random.Random(seed)
    cols = ["c%d" % i for i in range(FLIGHT_NCOLS)]
    lines = [",".join(cols)]
    for _ in range(n):
        cells = []
        for k in range(FLIGHT_NCOLS):
            m = k % 5
            if m == 0:
                cells.append(str(rng.randint(0, 100000)))
            elif m == 1:
                cells.append("%.2f" % rng.uniform(-500, 4000))
            elif m == 2:
                cells.append(rng.choice(["AA", "DL", "UA", "WN", "B6", "NK"]))
            elif m == 3:
                cells.append("City-%d Airport" % rng.randint(1, 300))
            else:
                cells.append(rng.choice(["on-time", "delayed", "cancelled", ""]))
        if rng.random() < bad_frac:
            if rng.random() < 0.5:
                cells = cells[:rng.randint(50, FLIGHT_NCOLS - 1)]  # underrun
            else:
                cells[0] = "notanint"
        lines.append(",".join(cells))
    return ("\n".join(lines) + "\n").encode()


def fl_ratio(x):
    return x["c1"] / (x["c0"] + 1)


def fl_carrier(x):
    return x["c2"] == "AA" or x["c2"] == "DL"


def fl_code(x):
    return x["c2"].lower()


def flights_ops():
    return [
        ("withColumn", "delay_ratio", fl_ratio),
        ("filter", fl_carrier),
        ("withColumn", "code", fl_code),
        ("selectColumns", ["c0", "code", "delay_ratio", "c3"]),
    ]


def make_weblog_lines(n, seed=3, bad_frac=0.01):
    """Space-tokenized weblog-ish lines (single str column for the engine; the
    logs config's string-split pipeline, BASELINE configs[4])."""
    rng = random.Random(seed)
    lines = []
    for _ in range(n):
        ip = "%d.%d.%d.%d" % (rng.randint(1, 255), rng.randint(0, 255),
                              rng.randint(0, 255), rng.randint(0, 255))
        method = rng.choice(["GET", "POST", "PUT"])
        url = "/page/%d.html" % rng.randint(1, 99999)
        code = rng.choice([200, 200, 200, 301, 404, 500])
        size = rng.randint(100, 2000000)
        if rng.random() < bad_frac:
            lines.append("%s %s %s" % (ip, method, url))  # short line
        else:
            lines.append("%s %s %s %d %d" % (ip, method, url, code, size))
    return ("\n".join(lines) + "\n").encode()


def log_parse(x):
    return (x.split(" ")[0], x.split(" ")[2], int(x.split(" ")[3]),
            int(x.split(" ")[4]))


def log_ok(x):
    return x[2] == 200


def logs_ops():
    return [
        ("map", log_parse),
        ("filter", log_ok),
    ]


def make_long_rows(n=200, seed=9):
    """Rows whose 64-row spans exceed the LDS staging cap (global-parse path)."""
    rng = random.Random(seed)
    vals = []
    for i in range(n):
        blob = "".join(rng.choice("abcdefgh ,x") for _ in range(rng.randint(500, 30000)))
        vals.append(blob)
    return vals
