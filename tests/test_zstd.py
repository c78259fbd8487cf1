"""zstd codec tests.

The codec is hand-written (core/include/modelx/zstd_core.hpp + zstd_enc.hpp,
shared verbatim with the CDNA4 kernels in core/hip/zstd.hip). CPU tests here
verify it against libzstd (dlopen'd system library) in BOTH directions — our
encoder must produce frames stock zstd decodes, and our decoder must decode
stock-zstd output at several levels — plus the full push→pull loop with
+zstd blobs against the C++ modelxd. GPU tests (test_gpu_kernels.py) compare
the kernels against this same CPU path.

Reference parity: the reference only has sequential CPU gzip
(pkg/client/helper.go:19-22); the +zstd media type is an MI355X-native
addition.
"""
import ctypes
import os
import random

import pytest

from modelx_amd import _core
from modelx_amd.client import Client
from modelx_amd.config import ModelConfig
from modelx_amd.wire import types

from util_servers import start_modelxd_local


@pytest.fixture(scope="module")
def libzstd():
    z = ctypes.CDLL("libzstd.so.1")
    z.ZSTD_compressBound.restype = ctypes.c_size_t
    z.ZSTD_compress.restype = ctypes.c_size_t
    z.ZSTD_decompress.restype = ctypes.c_size_t
    z.ZSTD_isError.restype = ctypes.c_uint
    return z


def _zstd_compress(z, data: bytes, level: int) -> bytes:
    bound = z.ZSTD_compressBound(len(data))
    dst = ctypes.create_string_buffer(bound)
    n = z.ZSTD_compress(dst, bound, data, len(data), level)
    assert not z.ZSTD_isError(n)
    return dst.raw[:n]


def _zstd_decompress(z, blob: bytes, raw_len: int) -> bytes:
    out = ctypes.create_string_buffer(max(raw_len, 1))
    m = z.ZSTD_decompress(out, raw_len, blob, len(blob))
    assert not z.ZSTD_isError(m), f"libzstd rc {m}"
    return out.raw[:m]


def _payloads():
    rng = random.Random(1234)
    reps = bytearray()
    while len(reps) < 300_000:
        if rng.random() < 0.6 and len(reps) > 64:
            off = rng.randrange(1, min(len(reps), 60_000))
            ln = rng.randrange(4, 300)
            start = len(reps) - off
            for k in range(ln):
                reps.append(reps[start + k])
        else:
            reps.extend(rng.randbytes(rng.randrange(1, 50)))
    return {
        "empty": b"",
        "tiny": b"a",
        "text": b"the quick brown fox jumps over the lazy dog " * 4000,
        "random": rng.randbytes(200_000),
        "zeros": bytes(150_000),
        "repeats": bytes(reps),
        "low-entropy": bytes(rng.choice(b"aab") for _ in range(140_000)),
        # 64-symbol uniform: no LZ matches -> pure huffman-literal blocks
        "entropy-only": bytes(rng.randrange(64) for _ in range(300_000)),
    }


class TestCpuCodec:
    def test_roundtrip(self):
        for name, data in _payloads().items():
            blob = _core.zstd_compress_cpu(data, 128 << 10)
            assert _core.zstd_decompress_cpu(blob) == data, name
            assert _core.zstd_content_size(blob) == len(data), name

    def test_frame_sizes(self):
        data = os.urandom(300_000)
        for frame_raw in (64 << 10, 128 << 10, 512 << 10):
            blob = _core.zstd_compress_cpu(data, frame_raw)
            frames = _core.zstd_frames(blob)
            assert len(frames) == (len(data) + frame_raw - 1) // frame_raw
            # frames tile the decompressed space
            assert sum(f[3] for f in frames) == len(data)
            assert _core.zstd_decompress_cpu(blob) == data

    def test_compresses(self):
        data = b"modelx " * 100_000
        blob = _core.zstd_compress_cpu(data, 128 << 10)
        assert len(blob) < len(data) // 20

    def test_huffman_literals_engage(self):
        import random

        rng = random.Random(5)
        data = bytes(rng.randrange(64) for _ in range(400_000))  # 6 bits/byte
        blob = _core.zstd_compress_cpu(data, 128 << 10)
        # entropy bound is 0.75; huffman-literal blocks should land near it
        assert len(blob) < int(len(data) * 0.85), len(blob)
        assert _core.zstd_decompress_cpu(blob) == data

    def test_our_encoder_libzstd_decodes(self, libzstd):
        for name, data in _payloads().items():
            blob = _core.zstd_compress_cpu(data, 128 << 10)
            assert _zstd_decompress(libzstd, blob, len(data)) == data, name

    def test_libzstd_encoder_we_decode(self, libzstd):
        for level in (1, 3, 9, 19):
            for name, data in _payloads().items():
                blob = _zstd_compress(libzstd, data, level)
                assert _core.zstd_decompress_cpu(blob) == data, (name, level)

    def test_corrupt_frame_detected(self):
        data = b"hello world " * 20_000
        blob = bytearray(_core.zstd_compress_cpu(data, 128 << 10))
        # corrupt inside the first frame's payload (past the 6-byte header)
        frames = _core.zstd_frames(bytes(blob))
        c_off, c_size = frames[0][0], frames[0][1]
        for flip in range(8, int(c_size) - 1, 7):
            mut = bytearray(blob)
            mut[c_off + flip] ^= 0xFF
            try:
                out = _core.zstd_decompress_cpu(bytes(mut))
            except RuntimeError:
                continue  # structural corruption detected
            # parseable corruption must change the content (the registry's
            # digest verification catches it one layer up)
            assert out != data, f"flip at {flip} silently decoded to identical bytes"

    def test_truncated_blob_raises(self):
        data = os.urandom(50_000)
        blob = _core.zstd_compress_cpu(data, 128 << 10)
        with pytest.raises(RuntimeError):
            _core.zstd_decompress_cpu(blob[: len(blob) // 2])


class TestZstdPushPull:
    @pytest.fixture(scope="class")
    def server(self, tmp_path_factory):
        p = start_modelxd_local(str(tmp_path_factory.mktemp("zstd-registry")))
        yield p
        p.stop()

    def test_push_pull_compressed(self, server, tmp_path):
        d = tmp_path / "model"
        d.mkdir()
        cfg = ModelConfig(description="zstd model")
        (d / "modelx.yaml").write_text(cfg.to_yaml())
        payload = b"weights " * 300_000 + os.urandom(64 << 10)
        (d / "weights.bin").write_bytes(payload)
        c = Client(server.url)
        manifest = c.push("lib/zstd-model", "v1", str(d), quiet=True, compress="zstd")
        (blob_desc,) = [b for b in manifest.blobs if b.name == "weights.bin"]
        assert blob_desc.media_type == types.MEDIA_TYPE_MODEL_FILE_ZSTD
        assert blob_desc.size < len(payload) // 2  # stored compressed
        assert blob_desc.annotations[types.ANNOTATION_RAW_SIZE] == str(len(payload))

        out = tmp_path / "out"
        c.pull("lib/zstd-model", "v1", str(out), quiet=True)
        assert (out / "weights.bin").read_bytes() == payload

    def test_pull_detects_corrupt_uncompressed_annotation(self, server, tmp_path):
        # push with a bad raw digest: pull must fail, not silently accept
        d = tmp_path / "model"
        d.mkdir()
        (d / "modelx.yaml").write_text(ModelConfig(description="x").to_yaml())
        (d / "w.bin").write_bytes(b"abc" * 50_000)
        c = Client(server.url)
        from modelx_amd.client.push import parse_manifest

        manifest = parse_manifest(str(d), compress="zstd")
        (blob,) = [b for b in manifest.blobs if b.name == "w.bin"]
        blob.annotations[types.ANNOTATION_RAW_DIGEST] = "sha256c1m:" + "0" * 64
        src = str(d / ".modelx" / "w.bin.zst")
        c.pusher.push_blob("lib/zstd-bad", blob, src)
        c.pusher.push_blob("lib/zstd-bad", manifest.config, str(d / "modelx.yaml"))
        c.remote.put_manifest("lib/zstd-bad", "v1", manifest)
        with pytest.raises(Exception):
            c.pull("lib/zstd-bad", "v1", str(tmp_path / "out2"), quiet=True)


def test_decoder_mutation_fuzz():
    """Bounded fuzz: random byte mutations of valid blobs must never
    crash the decoder — every outcome is either a clean error or a
    decode whose corruption the digest layer would catch."""
    rng = random.Random(31337)
    base = _payloads()
    for name in ("text", "repeats", "entropy-only"):
        data = base[name]
        blob = bytearray(_core.zstd_compress_cpu(data, 64 << 10))
        for _ in range(80):
            mut = bytearray(blob)
            for _ in range(rng.randrange(1, 4)):
                mut[rng.randrange(len(mut))] ^= rng.randrange(1, 256)
            try:
                _core.zstd_decompress_cpu(bytes(mut))
            except RuntimeError:
                pass  # clean rejection


def test_fse_weight_huffman_on_tensor_bytes():
    """Full-byte alphabets (bf16/fp32 tensor bytes carry the sign bit) use
    the FSE-compressed huffman weight table — before it, such payloads fell
    back to RAW blocks (ratio 1.000 vs 0.78 for zstd -3 on bf16 weights).
    Both our decoder and stock libzstd must decode the output."""
    import ctypes

    torch = pytest.importorskip("torch")
    z = ctypes.CDLL("libzstd.so.1")
    z.ZSTD_decompress.restype = ctypes.c_size_t
    z.ZSTD_isError.restype = ctypes.c_uint
    torch.manual_seed(3)
    cases = {
        "bf16": ((torch.randn(2 << 20) * 0.02).to(torch.bfloat16)
                 .view(torch.uint8).numpy().tobytes(), 0.85),
        "int8": ((torch.randn(1 << 20) * 30).clamp(-127, 127).to(torch.int8)
                 .view(torch.uint8).numpy().tobytes(), 0.95),
    }
    for name, (data, max_ratio) in cases.items():
        blob = _core.zstd_compress_cpu(data, 128 << 10)
        assert len(blob) / len(data) < max_ratio, name
        assert _core.zstd_decompress_cpu(blob) == data, name
        out = ctypes.create_string_buffer(len(data))
        m = z.ZSTD_decompress(out, len(data), blob, len(blob))
        assert not z.ZSTD_isError(m) and out.raw[:m] == data, name
