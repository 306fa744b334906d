"""API-surface checks (CPU): the drop-in Context/DataSet methods the reference
exposes (python/tuplex/dataset.py, context.py) exist with compatible
signatures and behave sensibly on the lazy side (no GPU needed — nothing here
executes a pipeline)."""
import inspect

import pytest

import tuplex_amd


def test_dataset_method_surface():
    ds_methods = ["map", "filter", "withColumn", "mapColumn", "selectColumns",
                  "renameColumn", "resolve", "ignore", "aggregate",
                  "aggregateByKey", "unique", "join", "leftJoin", "cache",
                  "collect", "take", "show", "tocsv", "toorc"]
    for m in ds_methods:
        assert hasattr(tuplex_amd.DataSet, m), m


def test_context_method_surface():
    ctx_methods = ["parallelize", "csv", "text", "orc", "options",
                   "optionsToYAML", "uiWebURL"]
    for m in ctx_methods:
        assert hasattr(tuplex_amd.Context, m), m


def test_join_signature_matches_reference():
    sig = inspect.signature(tuplex_amd.DataSet.join)
    assert list(sig.parameters)[:6] == ["self", "dsRight", "leftKeyColumn",
                                        "rightKeyColumn", "prefixes",
                                        "suffixes"]


def test_lazy_chaining_tracks_columns():
    ctx = tuplex_amd.Context()
    ds = ctx.parallelize([(1, "a")], columns=["k", "v"])
    ds2 = ds.withColumn("w", lambda x: x["k"] + 1).renameColumn("v", "val")
    assert ds2.columns == ["k", "val", "w"]
    ds3 = ds2.selectColumns(["val", "k"])
    assert ds3.columns == ["val", "k"]
    # chaining never mutates the parent
    assert ds.columns == ["k", "v"]


def test_options_surface_and_yaml(tmp_path):
    import os
    ctx = tuplex_amd.Context({"tuplex.partitionSize": "8MB"})
    o = ctx.options()
    assert o["tuplex.partitionSize"] == "8MB"
    for k in ["tuplex.inputSplitSize", "tuplex.normalcaseThreshold",
              "tuplex.optionalThreshold"]:
        assert k in o, k
    p = os.path.join(str(tmp_path), "conf.yaml")
    ctx.optionsToYAML(p)
    import yaml
    back = yaml.safe_load(open(p))
    assert back["tuplex.partitionSize"] == "8MB"


def test_join_requires_named_columns():
    ctx = tuplex_amd.Context()
    left = ctx.parallelize([(1, "a")], columns=["k", "v"])
    right = ctx.parallelize([(1, "x")])  # no names
    with pytest.raises(ValueError, match="named columns"):
        left.join(right, "k", "k")
