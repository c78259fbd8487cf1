"""Chunk-level resume planning (CPU; pure logic in GpuClient)."""
import hashlib

from modelx_amd.client.gpu import GpuClient


def _leaves(data: bytes, cs: int) -> bytes:
    return b"".join(hashlib.sha256(data[o : o + cs]).digest() for o in range(0, len(data), cs))


class TestBadChunkRanges:
    def test_no_mismatch(self):
        data = bytes(1000)
        lv = _leaves(data, 256)
        assert GpuClient._bad_chunk_ranges(lv, lv, 256, 1000) == []

    def test_single_bad_chunk(self):
        good = bytearray(1000)
        bad = bytearray(good)
        bad[300] ^= 1
        ranges = GpuClient._bad_chunk_ranges(_leaves(bytes(bad), 256), _leaves(bytes(good), 256),
                                             256, 1000)
        assert ranges == [(256, 256)]

    def test_adjacent_bad_chunks_merge(self):
        good = bytearray(1000)
        bad = bytearray(good)
        bad[260] ^= 1
        bad[600] ^= 1
        ranges = GpuClient._bad_chunk_ranges(_leaves(bytes(bad), 256), _leaves(bytes(good), 256),
                                             256, 1000)
        assert ranges == [(256, 512)]

    def test_tail_chunk_clamped(self):
        good = bytearray(1000)
        bad = bytearray(good)
        bad[999] ^= 1
        ranges = GpuClient._bad_chunk_ranges(_leaves(bytes(bad), 256), _leaves(bytes(good), 256),
                                             256, 1000)
        assert ranges == [(768, 232)]
