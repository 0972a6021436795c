"""Murmur3 hash tests: known vectors + cross-checks between the scalar
bytes path and the vectorized int paths (independent implementations)."""

import numpy as np
import pytest
import torch

from bigslice_amd import hashing
from bigslice_amd.frame import Frame


# Known murmur3_x86_32 test vectors (public algorithm).
KNOWN = [
    (b"", 0, 0),
    (b"", 1, 0x514E28B7),
    (b"hello", 0, 0x248BFA47),
    (b"hello, world", 0, 0x149BBB7F),
    (b"The quick brown fox jumps over the lazy dog", 0x9747B28C,
     0x2FA826CD),
]


def test_known_vectors():
    for data, seed, want in KNOWN:
        assert hashing.murmur3_bytes(data, seed) == want, data


def test_u32_matches_bytes():
    rng = np.random.default_rng(0)
    vals = rng.integers(0, 2**32, size=100, dtype=np.uint32)
    for seed in (0, 0x9ACB0442):
        vec = hashing.murmur3_u32(vals, seed)
        for i, v in enumerate(vals):
            b = int(v).to_bytes(4, "little")
            assert vec[i] == hashing.murmur3_bytes(b, seed)


def test_u64_matches_bytes():
    rng = np.random.default_rng(1)
    vals = rng.integers(0, 2**64, size=100, dtype=np.uint64)
    for seed in (0, 7):
        vec = hashing.murmur3_u64(vals, seed)
        for i, v in enumerate(vals):
            b = int(v).to_bytes(8, "little")
            assert vec[i] == hashing.murmur3_bytes(b, seed)


def test_signed_int_reinterpreted():
    # Go's uint32(int8(-1)) sign-extends; check our narrow-int path.
    t = torch.tensor([-1, 0, 127, -128], dtype=torch.int8)
    h = hashing._hash_host_column(t, 0)
    want0 = hashing.murmur3_bytes((0xFFFFFFFF).to_bytes(4, "little"), 0)
    assert h[0] == want0


def test_frame_hash_xor_combined():
    f = Frame([torch.tensor([1, 2, 3], dtype=torch.int64),
               torch.tensor([10, 20, 30], dtype=torch.int64),
               torch.tensor([0.5, 1.5, 2.5], dtype=torch.float64)],
              prefix=2)
    h = f.hash(0)
    h0 = hashing._hash_host_column(f.columns[0], 0)
    h1 = hashing._hash_host_column(f.columns[1], 0)
    assert (h.numpy() == (h0 ^ h1).astype(np.int64)).all()


def test_bool_hash():
    t = torch.tensor([True, False], dtype=torch.bool)
    h = hashing._hash_host_column(t, 5)
    assert h[0] == 6 and h[1] == 5


def test_string_column_hash():
    h = hashing._hash_host_column(["hello", "x"], 0)
    assert h[0] == 0x248BFA47


@pytest.mark.gpu
def test_device_hash_matches_host():
    from bigslice_amd import kernels
    assert kernels.have_extension(), "HIP extension must be built"
    for dt in (torch.int64, torch.int32, torch.float64, torch.float32):
        if dt.is_floating_point:
            t = torch.randn(100_000, dtype=dt)
        else:
            t = torch.randint(-2**31, 2**31 - 1, (100_000,), dtype=dt)
        host = hashing.hash_columns([t], 0)
        dev = hashing.hash_columns([t.cuda()], 0)
        assert (dev.cpu().to(torch.int64) == host.to(torch.int64)).all()
