"""Guard tests (CPU container): the normal-case path must FAIL LOUDLY without a
GPU (no silent CPU fallback), and the C baseline must agree with the oracle."""
import json
import os
import subprocess

import pytest

import tuplex_amd
from tuplex_amd.engine import GpuLib

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)


def gpu_present():
    try:
        return GpuLib.get().device_count() > 0
    except RuntimeError:
        return False


@pytest.mark.skipif(gpu_present(), reason="only meaningful without a GPU")
def test_compilable_pipeline_fails_loudly_without_gpu():
    from tests.pipelines import sq_map
    ds = tuplex_amd.Context().parallelize([1, 2, 3]).map(sq_map)
    with pytest.raises(RuntimeError, match="HIP|GPU"):
        ds.collect()


def test_czillow_matches_oracle_row_counts():
    """The cpu_baseline C port must produce exactly the oracle's output rows on
    the same input (it is a restatement, not just a timer)."""
    from oracle import pyoracle_csv
    from tests.test_codegen_compile import zillow_ops
    from tests.zillow_data import make_zillow_csv_bytes

    binp = os.path.join(REPO, "oracle", "czillow")
    if not os.path.exists(binp):
        subprocess.check_call(["make", "-C", os.path.join(REPO, "oracle"), "-s"])
    data, _ = make_zillow_csv_bytes(5000, seed=99, dirty_frac=0.02)
    tmp = "/tmp/tpx_czillow_test.csv"
    with open(tmp, "wb") as f:
        f.write(data)
    out = json.loads(subprocess.check_output([binp, tmp, "0"]))
    ref = pyoracle_csv.run_csv_pipeline(data, zillow_ops())
    assert out["rows_out"] == len(ref["output"]) * out["passes"]
    # czillow counts parse/UDF failures as exceptions without replay
    assert out["exceptions"] >= 0


def test_options_surface():
    ctx = tuplex_amd.Context({"partitionSize": "16MB"})
    opts = ctx.options()
    assert opts["tuplex.partitionSize"] == "16MB"
    assert "tuplex.normalcaseThreshold" in opts
    import tempfile
    p = tempfile.mktemp(suffix=".yaml")
    ctx.optionsToYAML(p)
    assert os.path.exists(p)
