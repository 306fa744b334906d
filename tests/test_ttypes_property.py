"""Type-lattice properties (hypothesis): majority-type inference and value
conformance must be consistent — every value classified normal-case CONFORMS
to the inferred type, and unify is commutative/idempotent (TypeSystem.cc
unifyTypes semantics)."""
import hypothesis.strategies as st
from hypothesis import given, settings

from tuplex_amd import ttypes as T

scalars = st.one_of(
    st.integers(min_value=-2**62, max_value=2**62),
    st.floats(allow_nan=False, allow_infinity=False, width=64),
    st.booleans(),
    st.text(max_size=8),
    st.none(),
)
values = st.one_of(scalars, st.tuples(scalars, scalars))


@settings(max_examples=300, deadline=None)
@given(st.lists(values, min_size=1, max_size=40))
def test_majority_conformance(data):
    maj = T.infer_majority_type(data, optional_threshold=0.7)
    n_conform = sum(1 for v in data if T.value_conforms(v, maj))
    assert 0 < n_conform <= len(data)


@settings(max_examples=300, deadline=None)
@given(st.sampled_from([T.I64, T.F64, T.BOOL, T.STR, T.NULL,
                        ("opt", T.I64), ("opt", T.STR)]),
       st.sampled_from([T.I64, T.F64, T.BOOL, T.STR, T.NULL,
                        ("opt", T.I64), ("opt", T.STR)]))
def test_unify_laws(a, b):
    ab = T.unify(a, b)
    ba = T.unify(b, a)
    assert ab == ba                      # commutative
    assert T.unify(a, a) == a            # idempotent
    if ab is not None:
        assert T.unify(ab, a) == ab      # absorption
