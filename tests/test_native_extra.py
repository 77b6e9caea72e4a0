"""LZ4 codec host tests + pinned buffers."""
import os
import random

from curvine_amd import native


def test_lz4_roundtrip_host():
    rng = random.Random(3)
    for data in (b"", b"xyz" * 40000, os.urandom(200_000),
                 bytes(rng.choices(b"ab", k=150_000))):
        assert native.lz4_decompress(native.lz4_compress(data)) == data


def test_pinned_buffer():
    b = native.PinnedBuffer(4096)
    b.view[:5] = b"hello"
    assert bytes(b.view[:5]) == b"hello"
    assert b.ptr != 0
    b.close()
