"""Wire-compat of the service messages against the official protobuf runtime."""
import numpy as np

from pytensor_federated_amd.npproto.utils import ndarray_from_numpy
from pytensor_federated_amd.rpc import (
    GetLoadParams,
    GetLoadResult,
    InputArrays,
    OutputArrays,
    ROUTE_EVALUATE_STREAM,
)


def _service_pool():
    from google.protobuf import descriptor_pb2, descriptor_pool

    pool = descriptor_pool.DescriptorPool()
    nd = descriptor_pb2.FileDescriptorProto(
        name="npproto/ndarray.proto", package="npproto", syntax="proto3"
    )
    md = nd.message_type.add(name="ndarray")
    md.field.add(name="data", number=1, type=12, label=1)
    md.field.add(name="dtype", number=2, type=9, label=1)
    md.field.add(name="shape", number=3, type=3, label=3)
    md.field.add(name="strides", number=4, type=3, label=3)
    pool.Add(nd)

    svc = descriptor_pb2.FileDescriptorProto(name="service.proto", syntax="proto3")
    svc.dependency.append("npproto/ndarray.proto")
    for msg_name in ("InputArrays", "OutputArrays"):
        m = svc.message_type.add(name=msg_name)
        f = m.field.add(name="items", number=1, type=11, label=3)
        f.type_name = ".npproto.ndarray"
        m.field.add(name="uuid", number=2, type=9, label=1)
    svc.message_type.add(name="GetLoadParams")
    m = svc.message_type.add(name="GetLoadResult")
    m.field.add(name="n_clients", number=1, type=5, label=1)
    m.field.add(name="percent_cpu", number=2, type=2, label=1)
    m.field.add(name="percent_ram", number=3, type=2, label=1)
    pool.Add(svc)
    return pool


def _cls(pool, name):
    from google.protobuf import message_factory

    return message_factory.GetMessageClass(pool.FindMessageTypeByName(name))


def test_input_arrays_wire_compat():
    pool = _service_pool()
    cls = _cls(pool, "InputArrays")
    a = np.array([1.0, 2.0])
    b = np.array(3.5)
    msg = InputArrays(items=[ndarray_from_numpy(a), ndarray_from_numpy(b)], uuid="abc-123")
    blob = bytes(msg)
    g = cls()
    g.ParseFromString(blob)
    assert g.uuid == "abc-123"
    assert len(g.items) == 2
    assert g.items[0].dtype == "float64" and g.items[0].data == a.tobytes()
    assert g.SerializeToString() == blob  # byte-identical round trip
    back = InputArrays().parse(g.SerializeToString())
    assert back.uuid == "abc-123" and len(back.items) == 2


def test_output_arrays_roundtrip():
    a = np.random.rand(4)
    msg = OutputArrays(items=[ndarray_from_numpy(a)], uuid="u")
    back = OutputArrays().parse(bytes(msg))
    assert back.uuid == "u"
    np.testing.assert_array_equal(
        np.frombuffer(back.items[0].data, dtype=np.float64), a
    )


def test_get_load_result_wire_compat():
    pool = _service_pool()
    cls = _cls(pool, "GetLoadResult")
    msg = GetLoadResult(n_clients=3, percent_cpu=12.5, percent_ram=40.25)
    g = cls()
    g.ParseFromString(bytes(msg))
    assert g.n_clients == 3
    assert abs(g.percent_cpu - 12.5) < 1e-6
    assert abs(g.percent_ram - 40.25) < 1e-6
    assert g.SerializeToString() == bytes(msg)
    back = GetLoadResult().parse(g.SerializeToString())
    assert back.n_clients == 3


def test_get_load_params_empty():
    assert bytes(GetLoadParams()) == b""
    GetLoadParams().parse(b"")


def test_route_matches_grpclib_convention():
    # no proto package -> the route the reference's grpclib stub uses
    # (reference service.py:231-236)
    assert ROUTE_EVALUATE_STREAM == "/ArraysToArraysService/EvaluateStream"
