"""Property-based robustness: serialization and encryption round-trips
hold for arbitrary payloads (hypothesis)."""
import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st

from vantage6_amd.common.encryption import RSACryptor
from vantage6_amd.common.serialization import deserialize, serialize

json_scalars = st.one_of(
    st.none(), st.booleans(),
    st.integers(min_value=-2**53, max_value=2**53),
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=40),
)
payloads = st.recursive(
    json_scalars,
    lambda inner: st.one_of(
        st.lists(inner, max_size=5),
        st.dictionaries(st.text(max_size=10), inner, max_size=5)),
    max_leaves=20,
)


@settings(max_examples=60, deadline=None)
@given(payloads)
def test_serialize_roundtrip(value):
    assert deserialize(serialize(value)) == value


@settings(max_examples=20, deadline=None)
@given(st.lists(st.floats(allow_nan=False, allow_infinity=False, width=32),
                min_size=1, max_size=50))
def test_ndarray_roundtrip(values):
    arr = np.array(values, dtype=np.float32)
    out = deserialize(serialize(arr))
    assert isinstance(out, np.ndarray)
    np.testing.assert_array_equal(out, arr)


_CRYPTOR = None


@settings(max_examples=25, deadline=None)
@given(st.binary(min_size=0, max_size=4096))
def test_hybrid_encryption_roundtrip(blob):
    # keygen is slow — one keypair for every generated example
    global _CRYPTOR
    if _CRYPTOR is None:
        import tempfile
        from pathlib import Path

        _CRYPTOR = RSACryptor(str(Path(tempfile.mkdtemp()) / "p.pem"),
                              bits=1024)
    enc = _CRYPTOR.encrypt_bytes_to_str(blob, _CRYPTOR.public_key_str)
    assert _CRYPTOR.decrypt_str_to_bytes(enc) == blob


@settings(max_examples=40, deadline=None)
@given(
    st.lists(st.floats(min_value=-1e3, max_value=1e3, allow_nan=False,
                       width=32), min_size=1, max_size=32),
    st.integers(min_value=2, max_value=5),
    st.integers(min_value=0, max_value=2**31),
)
def test_secagg_masks_cancel_exactly(vals, world, seed):
    """Pairwise int64 masks cancel bit-exactly under summation for any
    world size / payload / round seed, and every single rank's masked
    vector differs from its unmasked quantization (it reveals nothing)."""
    import torch

    from vantage6_amd.ops.secagg import quantize_mask

    scale = float(1 << 16)
    master = torch.tensor(vals, dtype=torch.float32)
    unmasked = (master.double() * scale).round().to(torch.int64)
    total = torch.zeros_like(unmasked)
    any_masked_differs = False
    for rank in range(world):
        q = torch.empty_like(unmasked)
        quantize_mask(q, master, 1.0, scale, seed=seed, rank=rank,
                      world_size=world)
        any_masked_differs |= bool((q != unmasked).any())
        total = total + q  # int64 wraps mod 2^64 — by design
    assert torch.equal(total, unmasked * world)
    assert any_masked_differs
