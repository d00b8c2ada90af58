"""Property-based fuzzing of the hand-written proto3 codec (hypothesis)."""
import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st

from pytensor_federated_amd.npproto import Ndarray
from pytensor_federated_amd.npproto.utils import ndarray_from_numpy, ndarray_to_numpy
from pytensor_federated_amd.proto_wire import (
    decode_varint,
    encode_varint,
    int64_from_uint,
)
from pytensor_federated_amd.rpc import GetLoadResult, InputArrays


@given(st.integers(min_value=0, max_value=(1 << 64) - 1))
def test_varint_roundtrip(v):
    blob = encode_varint(v)
    out, pos = decode_varint(blob, 0)
    assert out == v and pos == len(blob)


@given(st.integers(min_value=-(1 << 63), max_value=(1 << 63) - 1))
def test_int64_twos_complement_roundtrip(v):
    assert int64_from_uint(v & ((1 << 64) - 1)) == v


@given(
    st.binary(max_size=200),
    st.text(max_size=20),
    st.lists(st.integers(min_value=-(1 << 40), max_value=1 << 40), max_size=6),
    st.lists(st.integers(min_value=-(1 << 40), max_value=1 << 40), max_size=6),
)
@settings(max_examples=200)
def test_ndarray_message_roundtrip(data, dtype, shape, strides):
    msg = Ndarray(data=data, dtype=dtype, shape=shape, strides=strides)
    back = Ndarray().parse(bytes(msg))
    assert back.data == data
    assert back.dtype == dtype
    assert back.shape == shape
    assert back.strides == strides


@given(st.binary(max_size=300))
@settings(max_examples=300)
def test_parser_never_crashes_on_garbage(blob):
    """Arbitrary bytes must either parse or raise ValueError -- never hang
    or throw something uncontrolled."""
    for cls in (Ndarray, InputArrays, GetLoadResult):
        try:
            cls().parse(blob)
        except ValueError:
            pass


@given(
    st.lists(
        st.lists(st.floats(allow_nan=False, allow_infinity=False, width=64),
                 min_size=0, max_size=8),
        min_size=0, max_size=4,
    ),
    st.text(max_size=36),
)
@settings(max_examples=100)
def test_input_arrays_numpy_roundtrip(arrays, uuid):
    np_arrays = [np.asarray(a, dtype=np.float64) for a in arrays]
    msg = InputArrays(items=[ndarray_from_numpy(a) for a in np_arrays], uuid=uuid)
    back = InputArrays().parse(bytes(msg))
    assert back.uuid == uuid
    assert len(back.items) == len(np_arrays)
    for item, orig in zip(back.items, np_arrays):
        np.testing.assert_array_equal(ndarray_to_numpy(item), orig)
