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
