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


class _StubZstdEngine:
    """Stub for the batched-decode orchestration: records call shapes."""

    def __init__(self):
        self.decode_calls = []
        self.leaves_calls = []

    def zstd_decompress_many(self, items):
        self.decode_calls.append(len(items))
        return [cap for (_, _, _, cap) in items]

    def sha256_chunk_leaves_many(self, items):
        self.leaves_calls.append(len(items))
        return [b"\x00" * 32 for _ in items]


def test_zstd_batched_wave_split(monkeypatch):
    """>64 GiB of raw output splits into bounded waves; every job is
    decoded exactly once and results keep their keys."""
    import torch

    from modelx_amd.wire import types

    g = GpuClient.__new__(GpuClient)
    g.engine = _StubZstdEngine()
    g.device = 0
    g.last_stats = []

    def fake_pull(repository, desc, verify=True, plan_entry=None):
        return torch.zeros(max(desc.size, 1), dtype=torch.uint8)

    g.pull_blob_to_device = fake_pull
    # tensors land on CPU in the stub (torch.empty(device=cuda) would need
    # a GPU) — patch torch.empty's device arg through a shim
    real_empty = torch.empty

    def cpu_empty(*a, **kw):
        kw.pop("device", None)
        return real_empty(*a, **kw)

    monkeypatch.setattr(torch, "empty", cpu_empty)
    g.ZSTD_WAVE_CAP = 8 << 20
    jobs = []
    raw = 3 << 20  # 3 MiB each, 8 MiB cap -> waves of 2
    for i in range(4):
        d = types.Descriptor(name=f"w{i}", media_type=types.MEDIA_TYPE_MODEL_FILE_ZSTD,
                             digest=f"sha256:{i:064x}", size=1024,
                             annotations={types.ANNOTATION_RAW_SIZE: str(raw)})
        jobs.append((f"k{i}", d, None))
    out = g._pull_zstd_batched("r/x", jobs, verify=False, parallel=1)
    assert set(out) == {"k0", "k1", "k2", "k3"}
    assert sum(g.engine.decode_calls) == 4
    assert len(g.engine.decode_calls) >= 2  # split into waves
