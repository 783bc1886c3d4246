"""Server load test (reference: tests/load_tests/test_load_on_server.py
— N concurrent mixed requests through the in-process harness)."""
import concurrent.futures
import time

from tests.test_orchestrator import client, sky_env  # noqa: F401


def test_concurrent_mixed_requests(client):
    from skypilot_amd.client import sdk
    sdk.get(sdk.launch({"name": "base", "run": "true"}, "t-load"),
            timeout=90)

    def one_status(_):
        return sdk.get(sdk.status(), timeout=60)

    def one_queue(_):
        return sdk.get(sdk.queue("t-load"), timeout=60)

    t0 = time.time()
    with concurrent.futures.ThreadPoolExecutor(max_workers=16) as ex:
        futs = []
        for i in range(24):
            futs.append(ex.submit(one_status if i % 2 else one_queue, i))
        results = [f.result(timeout=120) for f in futs]
    dt = time.time() - t0
    assert len(results) == 24
    assert all(r is not None for r in results)
    assert dt < 60, f"24 concurrent short requests took {dt:.1f}s"
    sdk.get(sdk.down("t-load"))
