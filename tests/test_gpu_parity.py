"""GPU parity tests: the product's GPU path (through the C-ABI) against the oracle
on the same seeded inputs — the parity tests proper (SURVEY.md §8c bar: bit-exact
for integer/string/index work). Every test asserts the engine really ran in GPU
mode: a silent fallback would void the parity claim.

Pipelines live in tests/pipelines.py (shared with __graft_entry__.build()'s kernel
pre-compilation).
"""
import pytest

import tuplex_amd
from oracle import pyoracle
from tests.pipelines import PIPELINES, apply_ops

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("name,data,columns,ops",
                         PIPELINES, ids=[p[0] for p in PIPELINES])
def test_pipeline_parity(name, data, columns, ops):
    ctx = tuplex_amd.Context({"webui.enable": False})
    ds = apply_ops(ctx.parallelize(data, columns=columns), ops)
    got = ds.collect()
    ref = pyoracle.run_pipeline(data, ops, columns=columns)
    assert got == ref["output"], (name, got[:10], ref["output"][:10])
    assert ds.exception_counts == ref["exception_counts"], name
    assert ds._last_outcome.mode == "gpu", (name, ds._last_outcome.fallback_reason)


def test_empty_input():
    ctx = tuplex_amd.Context()
    from tests.pipelines import sq_map
    ds = ctx.parallelize([]).map(sq_map)
    assert ds.collect() == []


def test_zillow_keeps_real_fraction():
    from tests.pipelines import PIPELINES as P
    name, data, columns, ops = [p for p in P if p[0] == "zillow_mem"][0]
    ref = pyoracle.run_pipeline(data, ops, columns=columns)
    assert len(ref["output"]) > 100
