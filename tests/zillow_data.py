"""Synthetic Zillow-shaped data generator (shared by tests and bench.py).

Shape per SURVEY.md §8d: the 10-column header of the reference fixture
tuplex/test/resources/zillow_dirty_sample.csv:1 (title, address, city, state,
postal_code, price, "facts and features", real estate provider, url, sales_date),
~195 B/row mean, seeded. A small dirty fraction exercises the exception path
(unparseable facts, empty city) like the reference's dirty-zillow benchmark.
"""
import random

ZILLOW_COLS = ["title", "address", "city", "state", "postal_code", "price",
               "facts and features", "real estate provider", "url", "sales_date"]

_CITIES = ["WOBURN", "boston", "CAMBRIDGE", "Somerville", "medford", "QUINCY",
           "newton", "BROOKLINE", "arlington", "WALTHAM"]
_STATES = ["MA", "NY", "CA", "TX", "WA"]
_STREETS = ["Parker St", "Burlington Ave", "Main St", "Highland Rd", "Oak Dr",
            "Maple Ave", "Washington Blvd", "Elm St", "Cedar Ln", "Pine Ct"]
_PROVIDERS = ["J. Mulkerin Realty", "RE/MAX Destiny", "Redfin Corp", "",
              "Coldwell Banker", "Keller Williams", "Berkshire Hathaway"]
_KINDS = [("House For Sale", "sale", "house"),
          ("Condo For Sale", "sale", "condo"),
          ("House For Rent", "rent", "house"),
          ("Apartment For Rent", "rent", "condo"),
          ("House Sold", "sold", "house"),
          ("Condo Sold", "sold", "condo"),
          ("House Foreclosure", "foreclose", "house"),
          ("Land For Sale", "sale", "unknown")]


def _row(rng, dirty_frac):
    title, offer, _typ = _KINDS[rng.randrange(len(_KINDS))]
    bd = rng.randint(1, 12)
    ba = rng.randint(1, 4)
    sqft = rng.randint(300, 9000)
    pps = rng.randint(50, 900)
    dirty = rng.random() < dirty_frac
    if dirty and rng.random() < 0.5:
        facts = "studio apartment availability unknown"
    else:
        facts = "%d bds , %d ba , %s sqft" % (bd, ba, format(sqft, ","))
        if offer == "sold":
            facts = "Price/sqft: $%d , %s" % (pps, facts)
    if offer == "rent":
        price = "$%s/mo" % format(rng.randint(600, 12000), ",")
    else:
        price = "$%s" % format(rng.randint(40000, 30000000), ",")
    city = rng.choice(_CITIES)
    if dirty and rng.random() < 0.3:
        city = ""
    addr = "%d %s" % (rng.randint(1, 999), rng.choice(_STREETS))
    postal = float(rng.randint(1001, 99950))
    url = ("https://www.zillow.com/homedetails/%s-%s-%05d/%d_zpid/"
           % (addr.replace(" ", "-"), city or "X", int(postal),
              rng.randint(10**7, 10**8)))
    sales_date = rng.choice(["Open: Sat. 11am-1pm", "Sold: 04/%02d/2019"
                             % rng.randint(1, 28), "", "Open: Sun. 2-4pm"])
    return (title, addr, city, rng.choice(_STATES), postal, price, facts,
            rng.choice(_PROVIDERS), url, sales_date)


def make_zillow_rows(n, seed=42, dirty_frac=0.02):
    rng = random.Random(seed)
    return [_row(rng, dirty_frac) for _ in range(n)]


def _csv_cell(v):
    if isinstance(v, float):
        s = repr(v)
    else:
        s = str(v)
    if any(c in s for c in ',"\n\r'):
        return '"' + s.replace('"', '""') + '"'
    return s


def make_zillow_csv_bytes(n, seed=42, dirty_frac=0.02, header=True):
    rows = make_zillow_rows(n, seed, dirty_frac)
    out = []
    if header:
        out.append(",".join(_csv_cell(c) for c in ZILLOW_COLS))
    for r in rows:
        out.append(",".join(_csv_cell(v) for v in r))
    return ("\n".join(out) + "\n").encode(), rows
