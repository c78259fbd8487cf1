"""Wire-canonical digest path + GPU push retry logic (CPU-testable parts).

The canonical sha256 chain (reference: pkg/client/push.go:149-161) runs on
the CPU's SHA-NI units via OpenSSL EVP inside the native extension; the GPU
variant streams D2H through the pinned ring into the same code. These tests
oracle the EVP code against hashlib and exercise the retry wrapper with a
stub engine (the transport-level fault injection runs on the GPU box in
test_gpu_kernels.py).
"""
import hashlib

import pytest

from modelx_amd.client.gpu import GpuClient
from modelx_amd.wire import errors as er


@pytest.mark.parametrize("size", [0, 1, 55, 64, 65, 1000, 65536, (1 << 20) + 17])
def test_sha256_host_matches_hashlib(size):
    from modelx_amd import _core
    import os

    data = os.urandom(size)
    assert _core.sha256_host(data) == hashlib.sha256(data).digest()


class _FlakyEngine:
    """Stub engine whose push_part_from_device fails N times, then succeeds."""

    def __init__(self, failures):
        self.failures = failures
        self.calls = []

    def push_part_from_device(self, url, method, headers, ptr, length):
        self.calls.append((url, ptr, length))
        if len(self.calls) <= self.failures:
            raise RuntimeError("injected: connection reset")
        return {"status": 200}


def _client_with_engine(engine):
    g = GpuClient.__new__(GpuClient)
    g.engine = engine
    return g


def test_push_part_retries_transient_failure():
    eng = _FlakyEngine(failures=2)
    g = _client_with_engine(eng)
    g._push_part_retrying({"url": "http://x/part1"}, 0x1000, 4096)
    assert len(eng.calls) == 3  # 2 failures + 1 success


def test_push_part_gives_up_after_three():
    eng = _FlakyEngine(failures=99)
    g = _client_with_engine(eng)
    with pytest.raises(RuntimeError):
        g._push_part_retrying({"url": "http://x/part1"}, 0x1000, 4096)
    assert len(eng.calls) == GpuClient.PART_RETRIES


def test_push_from_gpu_rejects_unknown_digest_mode():
    g = GpuClient.__new__(GpuClient)
    with pytest.raises(er.ModelxError):
        g.push_from_gpu("r/x", "v1", {}, digest_mode="sha512")
