"""On-node device codec: HBM->HBM cross-process transfer via hipIpc (GPU)."""
import multiprocessing

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _receiver(blob: bytes, result_q):
    try:
        import torch

        from pytensor_federated_amd.npproto import Ndarray
        from pytensor_federated_amd.npproto.device import (
            close_imported_handles,
            device_ndarray_to_torch,
        )

        nda = Ndarray().parse(blob)
        t = device_ndarray_to_torch(nda, device="cuda:0")
        torch.cuda.synchronize()
        result_q.put(("ok", t.float().cpu().numpy()))
        close_imported_handles()
    except Exception as ex:
        result_q.put(("err", repr(ex)))


@pytest.mark.timeout(300)
def test_device_ndarray_cross_process_roundtrip():
    from pytensor_federated_amd.npproto.device import DeviceArrayExporter

    src = torch.arange(4096, dtype=torch.float32, device="cuda:0") * 0.5
    exporter = DeviceArrayExporter()
    nda = exporter.export(src)
    assert nda.dtype == "hipipc/float32"
    blob = bytes(nda)

    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_receiver, args=(blob, q), daemon=True)
    proc.start()
    status, payload = q.get(timeout=240)
    proc.join(timeout=30)
    assert status == "ok", payload
    np.testing.assert_array_equal(payload, src.cpu().numpy())


@pytest.mark.timeout(120)
def test_device_export_metadata():
    from pytensor_federated_amd.npproto.device import DeviceArrayExporter, is_device_ndarray

    src = torch.randn(333, dtype=torch.bfloat16, device="cuda:0")
    exporter = DeviceArrayExporter()
    nda = exporter.export(src)
    assert is_device_ndarray(nda)
    assert nda.dtype == "hipipc/bfloat16"
    assert nda.shape == [333]
    assert len(nda.data) == 96  # 64B handle + offset + nbytes + reserved
    # repeated export packs at a new offset in the same region
    nda2 = exporter.export(src)
    assert nda2.data[:64] == nda.data[:64]
    assert nda2.data != nda.data


def _device_worker(port, result_q):
    """Worker serving a GPU matmul over the fast transport with device-array
    replies (HBM->HBM both directions)."""
    try:
        import asyncio

        import torch

        from pytensor_federated_amd.fastsock import start_fast_server_async
        from pytensor_federated_amd.service import ArraysToArraysService

        def matmul(A, B):
            # inputs arrive as CUDA tensors (device codec); stay on device
            assert isinstance(A, torch.Tensor) and A.is_cuda, type(A)
            return [A.float() @ B.float()]

        async def main():
            service = ArraysToArraysService(matmul, device_arrays=True)
            server = await start_fast_server_async(service, "127.0.0.1", port)
            result_q.put("up")
            async with server:
                await server.serve_forever()

        asyncio.run(main())
    except Exception as ex:
        result_q.put(f"err: {ex!r}")


@pytest.mark.timeout(300)
def test_device_arrays_through_transport():
    """Full on-node device transport: client CUDA tensors -> IPC handles ->
    worker computes on device -> IPC reply -> client CUDA tensor."""
    import torch

    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    port = 9671
    proc = ctx.Process(target=_device_worker, args=(port, q), daemon=True)
    proc.start()
    try:
        assert q.get(timeout=240) == "up"
        client = ArraysToArraysServiceClient(
            "127.0.0.1", port, transport="fast", device_arrays=True
        )
        A = torch.randn(256, 512, device="cuda:0")
        B = torch.randn(512, 128, device="cuda:0")
        (C,) = client.evaluate(A, B)
        assert isinstance(C, torch.Tensor) and C.is_cuda
        torch.cuda.synchronize()
        ref = A @ B
        assert torch.allclose(C, ref, rtol=1e-4, atol=1e-4)
        # repeated calls reuse the IPC mappings
        for _ in range(10):
            (C,) = client.evaluate(A, B)
        torch.cuda.synchronize()
        assert torch.allclose(C, ref, rtol=1e-4, atol=1e-4)
        del client
    finally:
        proc.terminate()
        proc.join(timeout=10)
