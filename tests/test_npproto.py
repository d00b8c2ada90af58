"""Codec round-trip + wire-compatibility tests.

Mirrors reference test_npproto.py:10-31 (round-trip over parametrized
arrays) and adds a google.protobuf cross-check standing in for the
reference's betterproto golden-blob compatibility.
"""
import numpy as np
import pytest

from pytensor_federated_amd.npproto import Ndarray
from pytensor_federated_amd.npproto.utils import (
    ndarray_from_numpy,
    ndarray_to_numpy,
)


@pytest.mark.parametrize(
    "arr",
    [
        np.arange(5),
        np.random.uniform(size=(2, 3)),
        np.array(5),
        np.array(5.0),
        np.array("hello", dtype=str),
        np.array(["hello", "world"]),
        np.datetime64("2024-01-15"),
        np.arange("2023-01", "2023-04", dtype="datetime64[M]"),
        np.random.rand(7).astype(np.float32),
        np.array([], dtype=np.float64),
    ],
)
def test_roundtrip(arr):
    arr = np.asarray(arr)
    nda = ndarray_from_numpy(arr)
    blob = bytes(nda)
    back = ndarray_to_numpy(Ndarray().parse(blob))
    assert back.dtype == arr.dtype
    np.testing.assert_array_equal(back, arr)


def test_roundtrip_object_array_in_process():
    # ragged object array: pointers only valid in-process
    # (reference test_npproto.py:20; README.md:30)
    arr = np.array([[1, 2], [3]], dtype=object)
    back = ndarray_to_numpy(Ndarray().parse(bytes(ndarray_from_numpy(arr))))
    assert back.dtype == arr.dtype
    assert back[0] == [1, 2] and back[1] == [3]


def test_noncontiguous_view_roundtrips():
    a = np.arange(12.0).reshape(3, 4)[:, ::-1]
    back = ndarray_to_numpy(Ndarray().parse(bytes(ndarray_from_numpy(a))))
    np.testing.assert_array_equal(back, a)


def _protoc_ndarray_cls():
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    fdp = descriptor_pb2.FileDescriptorProto(
        name="npproto_test/ndarray.proto", package="npproto", syntax="proto3"
    )
    md = fdp.message_type.add(name="ndarray")
    md.field.add(name="data", number=1, type=12, label=1)
    md.field.add(name="dtype", number=2, type=9, label=1)
    md.field.add(name="shape", number=3, type=3, label=3)
    md.field.add(name="strides", number=4, type=3, label=3)
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    return message_factory.GetMessageClass(pool.FindMessageTypeByName("npproto.ndarray"))


def test_wire_compat_with_protobuf_runtime():
    """Our hand codec must be byte-identical with the official runtime."""
    cls = _protoc_ndarray_cls()
    a = np.arange(6.0).reshape(2, 3)
    ours = bytes(ndarray_from_numpy(a))
    g = cls()
    g.ParseFromString(ours)
    assert g.dtype == "float64"
    assert list(g.shape) == [2, 3]
    assert list(g.strides) == [24, 8]
    assert g.data == a.tobytes()
    theirs = cls(
        data=a.tobytes(), dtype="float64", shape=[2, 3], strides=[24, 8]
    ).SerializeToString()
    assert theirs == ours
    np.testing.assert_array_equal(a, ndarray_to_numpy(Ndarray().parse(theirs)))


def test_parse_accepts_unpacked_repeated_and_unknown_fields():
    from pytensor_federated_amd.proto_wire import (
        encode_int64_field,
        encode_len_delimited,
    )

    # strides as unpacked varints (proto2-style encoders) + unknown field 9
    blob = (
        encode_len_delimited(1, b"\x00" * 8)
        + encode_len_delimited(2, b"float64")
        + encode_int64_field(3, 1)
        + encode_int64_field(4, 8)
        + encode_len_delimited(9, b"future-extension")
    )
    nda = Ndarray().parse(blob)
    assert nda.shape == [1] and nda.strides == [8]
    arr = ndarray_to_numpy(nda)
    np.testing.assert_array_equal(arr, np.zeros(1))


def test_negative_strides_varint():
    # negative int64 encodes as 10-byte two's-complement varint
    nda = Ndarray(data=b"\x00" * 8, dtype="float64", shape=[1], strides=[-8])
    back = Ndarray().parse(bytes(nda))
    assert back.strides == [-8]
    cls = _protoc_ndarray_cls()
    g = cls()
    g.ParseFromString(bytes(nda))
    assert list(g.strides) == [-8]


def test_torch_tensor_roundtrip():
    import torch

    from pytensor_federated_amd.npproto.utils import (
        ndarray_from_torch,
        ndarray_to_torch,
    )

    for t in [
        torch.arange(6, dtype=torch.float32).reshape(2, 3),
        torch.randn(5, dtype=torch.float64),
        torch.randn(4, 3, dtype=torch.bfloat16),
    ]:
        blob = bytes(ndarray_from_torch(t))
        back = ndarray_to_torch(Ndarray().parse(blob))
        assert back.dtype == t.dtype
        assert torch.equal(back, t)
