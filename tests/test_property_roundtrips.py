"""Property-based robustness: serialization and encryption round-trips
hold for arbitrary payloads (hypothesis)."""
import numpy as np
from hypothesis import example, given, settings
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
@example(value={"__v6amd__": False})   # reserved-key collision (found by
@example(value={"__v6amd__": "ndarray", "nested": {"__v6amd__": 1}})  # fuzz
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


def test_totp_rfc6238_vectors():
    """RFC 6238 Appendix B SHA-1 test vectors (8 digits)."""
    import base64

    from vantage6_amd.common.totp import totp_code, verify_code

    secret = base64.b32encode(b"12345678901234567890").decode()
    for at, expect in [(59, "94287082"), (1111111109, "07081804"),
                       (1111111111, "14050471"), (1234567890, "89005924"),
                       (2000000000, "69279037")]:
        assert totp_code(secret, at=at, digits=8) == expect
    # 6-digit verify: current and ±1 window accepted, ±3 rejected
    now = 1700000000.0
    from vantage6_amd.common.totp import totp_code as tc
    assert verify_code(secret, tc(secret, at=now), at=now)
    assert verify_code(secret, tc(secret, at=now - 30), at=now)
    assert not verify_code(secret, tc(secret, at=now - 120), at=now)


@settings(max_examples=30, deadline=None)
@given(
    st.lists(st.sampled_from(["a", "b", "c", "d"]), min_size=1, max_size=60),
    st.integers(min_value=2, max_value=6),
)
def test_crosstab_station_never_leaks_small_cells(labels, threshold):
    """Disclosure control is enforced AT THE STATION: no emitted count is
    ever below the suppression threshold, and suppressed cells are
    counted but not named."""
    import pandas as pd

    from algorithms.crosstab import partial_counts

    df = pd.DataFrame({"g": labels})
    out = partial_counts.__wrapped__(df, "g", None,
                                     suppress_below=threshold, min_rows=1)
    assert all(v >= threshold for v in out["counts"].values())
    true_cells = df["g"].value_counts()
    assert out["suppressed_cells"] == int((true_cells < threshold).sum())


@settings(max_examples=30, deadline=None)
@given(st.lists(st.floats(min_value=-1e4, max_value=1e4, allow_nan=False,
                          width=32), min_size=1, max_size=64),
       st.sampled_from(["float32", "bfloat16", "int64"]))
def test_tensor_roundtrip_all_dtypes(vals, dtype):
    """Tagged tensor serialization round-trips bf16 (uint16 view), fp32
    and int64 exactly."""
    import torch

    t = torch.tensor(vals)
    if dtype == "int64":
        t = t.long()
    elif dtype == "bfloat16":
        t = t.to(torch.bfloat16)
    out = deserialize(serialize({"t": t}))["t"]
    assert out.dtype == t.dtype and out.shape == t.shape
    assert torch.equal(out, t)


@settings(max_examples=60, deadline=None)
@given(st.binary(min_size=0, max_size=200))
def test_deserialize_garbage_fails_cleanly(blob):
    """Adversarial bytes either decode to a value or raise a clean
    ValueError-family error — never KeyError/AttributeError/hang (stations
    deserialize attacker-supplied task inputs)."""
    try:
        deserialize(blob)
    except (ValueError, UnicodeDecodeError):  # json/np errors subclass these
        pass
    except Exception as exc:  # pragma: no cover
        from vantage6_amd.common.serialization import SerializationError

        assert isinstance(exc, SerializationError), repr(exc)


def test_deserialize_adversarial_tags_fail_cleanly():
    """Crafted tag payloads (object dtype, absurd shapes, truncated data,
    unknown tags) raise clean errors without huge allocations."""
    import json

    import pytest

    from vantage6_amd.common.serialization import SerializationError

    cases = [
        {"__v6amd__": "ndarray", "dtype": "object", "shape": [1], "data": "AA=="},
        {"__v6amd__": "ndarray", "dtype": "float32", "shape": [2**40],
         "data": "AAAA"},                      # shape >> buffer
        {"__v6amd__": "ndarray", "dtype": "not-a-dtype", "shape": [1],
         "data": "AAAA"},
        {"__v6amd__": "tensor", "dtype": "float32", "shape": [-3],
         "data": "AAAA"},
        {"__v6amd__": "totally-unknown", "data": ""},
        {"__v6amd__": "bytes", "data": "!!!notb64!!!"},
    ]
    for c in cases:
        with pytest.raises((ValueError, TypeError, SerializationError)):
            deserialize(json.dumps(c).encode())


_FUZZ_CRYPTOR = None


@settings(max_examples=40, deadline=None)
@given(st.text(max_size=120))
def test_decrypt_garbage_fails_cleanly(envelope):
    """Random envelope strings never crash decrypt with anything but
    ValueError (stations decrypt attacker-reachable inputs)."""
    global _FUZZ_CRYPTOR
    if _FUZZ_CRYPTOR is None:
        import tempfile
        from pathlib import Path

        _FUZZ_CRYPTOR = RSACryptor(
            str(Path(tempfile.mkdtemp()) / "fz.pem"), bits=1024)
    try:
        _FUZZ_CRYPTOR.decrypt_str_to_bytes(envelope)
    except ValueError:
        pass


def test_totp_corrupt_secret_returns_false():
    from vantage6_amd.common.totp import verify_code

    assert verify_code("not!!base32", "123456") is False
    assert verify_code("", "123456") is False
