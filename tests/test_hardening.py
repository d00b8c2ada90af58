"""Regression tests for round-2 transport/codec hardening.

Covers the advisor + judge findings from round 1:
* untrusted device-ndarray headers are validated before any device work,
* fast-transport frames are size-capped in both directions,
* grpc.aio stream end is detected structurally (not via repr() duck-typing),
* the connection cache is bounded,
* device-array exporters are per-connection, not shared across streams.
"""
import asyncio
import struct

import numpy as np
import pytest

import pytensor_federated_amd.service as service_mod
from pytensor_federated_amd.npproto import Ndarray
from pytensor_federated_amd.npproto.device import (
    DEVICE_DTYPE_PREFIX,
    _HEADER,
    device_ndarray_to_torch,
)
from pytensor_federated_amd.rpc import InputArrays, OutputArrays
from pytensor_federated_amd.service import (
    ArraysToArraysService,
    _privates,
    _streamed_evaluate,
)


def _device_nda(shape, dtype="float32", nbytes=None, offset=0, data=None):
    import torch

    if nbytes is None:
        numel = int(np.prod(shape)) if shape else 1
        nbytes = numel * torch.empty(0, dtype=getattr(torch, dtype)).element_size()
    payload = data if data is not None else _HEADER.pack(b"\x00" * 64, offset, nbytes)
    return Ndarray(
        data=payload,
        dtype=DEVICE_DTYPE_PREFIX + dtype,
        shape=list(shape),
        strides=[],
    )


class TestDeviceHeaderValidation:
    """A hostile wire header must be rejected before any device copy."""

    def test_nbytes_mismatch_rejected(self):
        nda = _device_nda([4, 4], nbytes=17)
        with pytest.raises(ValueError, match="does not match"):
            device_ndarray_to_torch(nda)

    def test_oversized_nbytes_rejected(self):
        # the original bug: nbytes from the wire > prod(shape)*itemsize
        # would have written past the destination tensor
        nda = _device_nda([8], nbytes=1 << 30)
        with pytest.raises(ValueError, match="does not match"):
            device_ndarray_to_torch(nda)

    def test_negative_offset_rejected(self):
        nda = _device_nda([8], offset=-64)
        with pytest.raises(ValueError, match="offset"):
            device_ndarray_to_torch(nda)

    def test_truncated_header_rejected(self):
        nda = _device_nda([8], data=b"short")
        with pytest.raises(ValueError, match="header"):
            device_ndarray_to_torch(nda)

    def test_negative_dim_rejected(self):
        numel_bytes = 8 * 4
        nda = _device_nda([-8], nbytes=numel_bytes)
        with pytest.raises(ValueError):
            device_ndarray_to_torch(nda)

    def test_unknown_dtype_rejected(self):
        nda = Ndarray(
            data=_HEADER.pack(b"\x00" * 64, 0, 8),
            dtype=DEVICE_DTYPE_PREFIX + "not_a_dtype",
            shape=[2],
            strides=[],
        )
        with pytest.raises(ValueError, match="dtype"):
            device_ndarray_to_torch(nda)


class TestFrameCap:
    def test_read_frame_rejects_oversized_announcement(self):
        from pytensor_federated_amd import fastsock

        async def run():
            reader = asyncio.StreamReader()
            hdr = bytes([fastsock.T_EVAL]) + (fastsock.MAX_FRAME_BYTES + 1).to_bytes(
                4, "little"
            )
            reader.feed_data(hdr)
            with pytest.raises(fastsock.FrameTooLargeError):
                await fastsock._read_frame(reader)

        asyncio.run(run())

    def test_frame_build_refuses_oversized_payload(self, monkeypatch):
        from pytensor_federated_amd import fastsock

        monkeypatch.setattr(fastsock, "MAX_FRAME_BYTES", 64)
        with pytest.raises(fastsock.FrameTooLargeError):
            fastsock._frame(fastsock.T_EVAL, b"x" * 65)

    def test_server_closes_connection_on_oversized_frame(self):
        """End-to-end: a hostile frame header closes the connection instead of
        allocating the announced payload."""
        from pytensor_federated_amd import fastsock

        async def run():
            service = ArraysToArraysService(lambda a: [a])
            server = await fastsock.start_fast_server_async(service, "127.0.0.1", 0)
            port = server.sockets[0].getsockname()[1]
            try:
                reader, writer = await asyncio.open_connection("127.0.0.1", port)
                writer.write(fastsock.MAGIC)
                writer.write(
                    bytes([fastsock.T_EVAL])
                    + (fastsock.MAX_FRAME_BYTES + 1).to_bytes(4, "little")
                )
                await writer.drain()
                # server must drop the connection without reading a payload
                got = await asyncio.wait_for(reader.read(), timeout=10)
                assert got == b""  # EOF
                writer.close()
            finally:
                server.close()
                await server.wait_closed()

        asyncio.run(run())


class TestStreamEofDetection:
    def test_grpc_eof_sentinel_raises_connectionerror(self):
        import grpc.aio

        class FakeStream:
            async def write(self, msg):
                pass

            async def read(self):
                return grpc.aio.EOF

        with pytest.raises(ConnectionError):
            asyncio.run(_streamed_evaluate(FakeStream(), InputArrays(items=[], uuid="u")))

    def test_none_raises_connectionerror(self):
        class FakeStream:
            async def write(self, msg):
                pass

            async def read(self):
                return None

        with pytest.raises(ConnectionError):
            asyncio.run(_streamed_evaluate(FakeStream(), InputArrays(items=[], uuid="u")))

    def test_real_message_passes(self):
        out = OutputArrays(items=[], uuid="u")

        class FakeStream:
            async def write(self, msg):
                pass

            async def read(self):
                return out

        got = asyncio.run(_streamed_evaluate(FakeStream(), InputArrays(items=[], uuid="u")))
        assert got is out


class TestBoundedPrivatesCache:
    def test_eviction_closes_oldest(self, monkeypatch):
        monkeypatch.setattr(service_mod, "_PRIVATES_MAX", 3)
        closed = []

        class FakeStream:
            def __init__(self, name):
                self.name = name

            def cancel(self):
                closed.append(self.name)

        before = dict(_privates)
        _privates.clear()
        try:

            async def run():
                for i in range(6):
                    _privates[f"cid{i}"] = service_mod.ClientPrivates(
                        None, FakeStream(f"cid{i}"), "h", i
                    )
                    await service_mod._evict_privates_lru()

            asyncio.run(run())
            assert len(_privates) == 3
            assert closed == ["cid0", "cid1", "cid2"]
            assert set(_privates) == {"cid3", "cid4", "cid5"}
        finally:
            _privates.clear()
            _privates.update(before)


class TestPerConnectionExporter:
    def test_each_stream_gets_its_own_exporter(self, monkeypatch):
        """Two concurrent streams must not share one export region."""

        class FakeExporter:
            instances = []

            def __init__(self):
                self.resets = 0
                FakeExporter.instances.append(self)

            def reset(self):
                self.resets += 1

        service = ArraysToArraysService(lambda a: [np.asarray(a)])
        service._device_arrays = True
        monkeypatch.setattr(
            ArraysToArraysService, "_new_exporter", lambda self: FakeExporter()
        )

        seen = []

        async def one_stream(n_requests):
            async def reqs():
                from pytensor_federated_amd.npproto.utils import ndarray_from_numpy

                for _ in range(n_requests):
                    yield InputArrays(
                        items=[ndarray_from_numpy(np.asarray(1.0))], uuid="u"
                    )
                    await asyncio.sleep(0.01)

            outs = []
            async for out in service.evaluate_stream(reqs()):
                outs.append(out)
            return outs

        async def run():
            await asyncio.gather(one_stream(3), one_stream(3))

        # _run_compute_func only passes exporter to CUDA outputs; here we just
        # assert the exporter identity/diversity and reset bookkeeping.
        asyncio.run(run())
        assert len(FakeExporter.instances) == 2
        assert [e.resets for e in FakeExporter.instances] == [3, 3]

    def test_unary_rotates_ring(self, monkeypatch):
        class FakeExporter:
            count = 0

            def __init__(self):
                FakeExporter.count += 1

            def reset(self):
                pass

        service = ArraysToArraysService(lambda a: [np.asarray(a)])
        service._device_arrays = True
        monkeypatch.setattr(
            ArraysToArraysService, "_new_exporter", lambda self: FakeExporter()
        )
        from pytensor_federated_amd.npproto.utils import ndarray_from_numpy

        async def run():
            for _ in range(10):
                await service.evaluate(
                    InputArrays(items=[ndarray_from_numpy(np.asarray(1.0))], uuid="u")
                )

        asyncio.run(run())
        # ring of 4 exporters total, reused round-robin
        assert FakeExporter.count == ArraysToArraysService._UNARY_RING


class TestEssPairing:
    """Stan/arviz pairing: iid chains give ESS ~ total draws; a strongly
    autocorrelated AR(1) chain matches tau = (1+phi)/(1-phi) to ~25%."""

    def test_iid_ess_near_total(self):
        from pytensor_federated_amd.inference.diagnostics import effective_sample_size

        rng = np.random.default_rng(42)
        chains = rng.normal(size=(4, 2000))
        ess = effective_sample_size(chains)
        assert 0.7 * 8000 <= ess <= 8000

    def test_ar1_ess_matches_theory(self):
        from pytensor_federated_amd.inference.diagnostics import effective_sample_size

        rng = np.random.default_rng(7)
        phi = 0.9
        n, m = 20000, 4
        chains = np.empty((m, n))
        for c in range(m):
            x = 0.0
            innov = rng.normal(size=n) * np.sqrt(1 - phi**2)
            for i in range(n):
                x = phi * x + innov[i]
                chains[c, i] = x
        tau_true = (1 + phi) / (1 - phi)  # = 19
        ess = effective_sample_size(chains)
        ess_true = m * n / tau_true
        assert 0.75 * ess_true <= ess <= 1.35 * ess_true
