import threading

import torch

from sheeprl_amd.data.prefetch import DevicePrefetcher


def test_prefetcher_cpu_passthrough():
    calls = []

    def sample():
        calls.append(1)
        return {"x": torch.full((4,), float(len(calls)))}

    pf = DevicePrefetcher(sample, torch.device("cpu"), depth=2)
    b1 = pf.next()
    b2 = pf.next()
    assert b1["x"][0] <= b2["x"][0]
    pf.close()


def test_prefetcher_lock_shared_with_writer():
    lock = threading.Lock()
    data = {"v": 0}

    def sample():
        return {"x": torch.tensor([float(data["v"])])}

    pf = DevicePrefetcher(sample, torch.device("cpu"), depth=1, lock=lock)
    for i in range(5):
        with lock:
            data["v"] = i
        pf.next()
    pf.close()
