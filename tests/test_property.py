"""Property-based tests (hypothesis) for the codec and digest layers —
identity and oracle properties over arbitrary inputs, beyond the
hand-picked cases in test_zstd.py / test_wire.py."""
import hashlib

import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from modelx_amd import _core
from modelx_amd.wire import digest as dg


@settings(max_examples=60, deadline=None)
@given(st.binary(min_size=0, max_size=300_000),
       st.sampled_from([4 << 10, 64 << 10, 128 << 10]))
def test_zstd_roundtrip_identity(data, frame):
    blob = _core.zstd_compress_cpu(data, frame)
    assert _core.zstd_decompress_cpu(blob) == data


@settings(max_examples=40, deadline=None)
@given(st.binary(min_size=1, max_size=100_000))
def test_zstd_libzstd_decodes_ours(data):
    import ctypes

    z = ctypes.CDLL("libzstd.so.1")
    z.ZSTD_decompress.restype = ctypes.c_size_t
    z.ZSTD_isError.restype = ctypes.c_uint
    blob = _core.zstd_compress_cpu(data, 64 << 10)
    out = ctypes.create_string_buffer(len(data))
    n = z.ZSTD_decompress(out, len(data), blob, len(blob))
    assert not z.ZSTD_isError(n) and out.raw[:n] == data


@settings(max_examples=60, deadline=None)
@given(st.binary(min_size=0, max_size=200_000), st.integers(1, 5))
def test_streaming_digester_matches_oneshot(data, pieces):
    d = dg.StreamingDigester(chunk_size=64 << 10)
    step = max(1, len(data) // pieces)
    for off in range(0, max(len(data), 1), step):
        d.update(data[off:off + step])
    assert d.canonical_digest() == dg.sha256_digest(data)
    assert d.chunk_digest() == dg.chunked_digest(data, 64 << 10)


@settings(max_examples=60, deadline=None)
@given(st.binary(min_size=0, max_size=100_000))
def test_sha256_host_oracle(data):
    assert _core.sha256_host(data) == hashlib.sha256(data).digest()


@settings(max_examples=40, deadline=None)
@given(st.binary(min_size=17, max_size=20_000))
def test_decoder_never_crashes_on_garbage(data):
    """Arbitrary bytes into the seekable decoder: clean error or decode,
    never a crash (the fuzz class that found the FSE state OOB)."""
    try:
        _core.zstd_decompress_cpu(data)
    except RuntimeError:
        pass
