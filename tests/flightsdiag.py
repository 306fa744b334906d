"""GPU-box diagnostic: which flights rows divert on the GPU vs the oracle."""
import sys

sys.path.insert(0, "/root/repo")
import tuplex_amd
from oracle import pyoracle_csv
from tests import extra_data as X
from tests.pipelines import apply_ops, csv_used_cols

data = X.make_flights_csv(3000, seed=7, bad_frac=0.01)
with open("/tmp/fl.csv", "wb") as f:
    f.write(data)
ctx = tuplex_amd.Context()
ds = apply_ops(ctx.csv("/tmp/fl.csv"), X.flights_ops())
got = ds.collect()
print("mode", ds._last_outcome.mode, "n_got", len(got))
ref = pyoracle_csv.run_csv_pipeline(
    data, X.flights_ops(), used_cols=csv_used_cols(data, X.flights_ops()))
print("n_ref", len(ref["output"]))
print("exc got", ds.exception_counts, "exc ref", ref["exception_counts"])
ndiff = 0
for i, (g, r) in enumerate(zip(got, ref["output"])):
    if g != r:
        ndiff += 1
        if ndiff <= 5:
            print("DIFF at", i, "got", g, "ref", r)
print("ndiff", ndiff, "len diff", len(got) - len(ref["output"]))
