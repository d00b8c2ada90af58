#!/usr/bin/env python3
"""Device-array transport vs host-byte transport for large on-node arrays."""
import asyncio
import json
import multiprocessing
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def _worker(port, device_arrays, q):
    import asyncio

    import torch

    from pytensor_federated_amd.fastsock import start_fast_server_async
    from pytensor_federated_amd.service import ArraysToArraysService

    def echo_scale(A):
        if isinstance(A, torch.Tensor):
            return [A * 2]
        return [torch.as_tensor(A, device="cuda:0") * 2]

    async def main():
        service = ArraysToArraysService(echo_scale, device_arrays=device_arrays)
        server = await start_fast_server_async(service, "127.0.0.1", port)
        q.put("up")
        async with server:
            await server.serve_forever()

    asyncio.run(main())


def main():
    import torch

    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    assert torch.cuda.is_available()
    results = {}
    for mb in (1, 16):
        n = mb * 1024 * 1024 // 4
        for mode in ("host", "device"):
            port = 9681 if mode == "host" else 9682
            ctx = multiprocessing.get_context("spawn")
            q = ctx.Queue()
            proc = ctx.Process(target=_worker, args=(port, mode == "device", q), daemon=True)
            proc.start()
            try:
                assert q.get(timeout=240) == "up"
                client = ArraysToArraysServiceClient(
                    "127.0.0.1", port, transport="fast", device_arrays=(mode == "device")
                )
                A = torch.randn(n, device="cuda:0")
                src = A if mode == "device" else A.cpu().numpy()
                for _ in range(5):
                    client.evaluate(src)
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                iters = 40
                for _ in range(iters):
                    client.evaluate(src)
                torch.cuda.synchronize()
                results[f"{mb}MB_{mode}"] = (time.perf_counter() - t0) / iters * 1000
                del client
            finally:
                proc.terminate()
                proc.join(timeout=10)
    print(json.dumps({"ms_per_roundtrip": results,
                      "note": "echo-scale of an fp32 array, on-node fast transport"}))


if __name__ == "__main__":
    main()
