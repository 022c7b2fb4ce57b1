"""End-to-end hybrid encryption between organizations.

Reference parity: ``vantage6-common/vantage6/common/encryption.py``
 — ``CryptorBase`` (base64 passthrough for unencrypted
collaborations), ``DummyCryptor`` and ``RSACryptor`` implementing a hybrid
scheme: a random shared key encrypts the payload with a symmetric cipher, and
the shared key is RSA-OAEP-encrypted with the recipient organization's public
key. Wire format: ``$``-joined base64 fields, as in the reference.

Deviation from the reference (documented, deliberate): this image has no
``cryptography`` wheel and no network, so the primitives are implemented
here directly —

- RSA-OAEP(SHA-256) on big integers (keygen via Miller-Rabin, e = 65537),
- the symmetric cipher is an HMAC-SHA256 counter-mode stream cipher with an
  encrypt-then-MAC tag instead of AES-CTR (same interface and security
  role; AES is unavailable without OpenSSL bindings).

Wire format (reference-shaped, 3 ``$``-joined fields like upstream's
``b64(enc_key)$b64(iv)$b64(ciphertext)``):

    ``b64(rsa(shared_key)) $ b64(nonce) $ b64(ct || tag)``

with the 32-byte encrypt-then-MAC tag folded into the tail of the third
field (upstream AES-CTR has no tag; the extra 32 bytes are this
implementation's authenticity deviation, visible only at this boundary).
Decryption also accepts the legacy round-1 4-field envelope
``... $ b64(ct) $ b64(tag)``.
"""
from __future__ import annotations

import hashlib
import hmac
import json
import os
import secrets
from pathlib import Path

from . import bytes_to_base64s, base64s_to_bytes

SEP = "$"
_HASH = hashlib.sha256
_HLEN = 32


class CryptorBase:
    """Base64 passthrough "cryptor" used for unencrypted collaborations."""

    def bytes_to_str(self, data: bytes) -> str:
        return bytes_to_base64s(data)

    def str_to_bytes(self, data: str) -> bytes:
        return base64s_to_bytes(data)

    def encrypt_bytes_to_str(self, data: bytes, pubkey_base64s: str | None) -> str:
        return self.bytes_to_str(data)

    def decrypt_str_to_bytes(self, data: str) -> bytes:
        return self.str_to_bytes(data)


DummyCryptor = CryptorBase


# --------------------------------------------------------------------------
# symmetric layer: HMAC-SHA256 CTR keystream + encrypt-then-MAC
# --------------------------------------------------------------------------
def _keystream_xor(key: bytes, nonce: bytes, data: bytes) -> bytes:
    out = bytearray(len(data))
    blocks = (len(data) + _HLEN - 1) // _HLEN
    pos = 0
    for ctr in range(blocks):
        block = hmac.new(key, nonce + ctr.to_bytes(8, "big"), _HASH).digest()
        n = min(_HLEN, len(data) - pos)
        for i in range(n):
            out[pos + i] = data[pos + i] ^ block[i]
        pos += n
    return bytes(out)


def _sym_encrypt(key: bytes, data: bytes) -> tuple[bytes, bytes, bytes]:
    nonce = secrets.token_bytes(16)
    enc_key = hmac.new(key, b"enc", _HASH).digest()
    mac_key = hmac.new(key, b"mac", _HASH).digest()
    ct = _keystream_xor(enc_key, nonce, data)
    tag = hmac.new(mac_key, nonce + ct, _HASH).digest()
    return nonce, ct, tag


def _sym_decrypt(key: bytes, nonce: bytes, ct: bytes, tag: bytes) -> bytes:
    enc_key = hmac.new(key, b"enc", _HASH).digest()
    mac_key = hmac.new(key, b"mac", _HASH).digest()
    expect = hmac.new(mac_key, nonce + ct, _HASH).digest()
    if not hmac.compare_digest(tag, expect):
        raise ValueError("symmetric-layer authentication failed")
    return _keystream_xor(enc_key, nonce, ct)


# --------------------------------------------------------------------------
# RSA on big integers
# --------------------------------------------------------------------------
_SMALL_PRIMES = [2, 3, 5, 7, 11, 13, 17, 19, 23, 29, 31, 37, 41, 43, 47, 53, 59, 61, 67]


def _is_probable_prime(n: int, rounds: int = 40) -> bool:
    if n < 2:
        return False
    for p in _SMALL_PRIMES:
        if n % p == 0:
            return n == p
    d, r = n - 1, 0
    while d % 2 == 0:
        d //= 2
        r += 1
    for _ in range(rounds):
        a = secrets.randbelow(n - 3) + 2
        x = pow(a, d, n)
        if x in (1, n - 1):
            continue
        for _ in range(r - 1):
            x = pow(x, 2, n)
            if x == n - 1:
                break
        else:
            return False
    return True


def _gen_prime(bits: int) -> int:
    while True:
        p = secrets.randbits(bits) | (1 << (bits - 1)) | 1
        if _is_probable_prime(p):
            return p


class RSAKey:
    """An RSA keypair (public part always present, private part optional)."""

    def __init__(self, n: int, e: int, d: int | None = None):
        self.n, self.e, self.d = n, e, d

    # -- generation / (de)serialization ------------------------------------
    @classmethod
    def generate(cls, bits: int = 2048) -> "RSAKey":
        if bits < 1024:
            # OAEP-SHA256 payload needs k >= 2*32+2 bytes + the shared key
            raise ValueError("RSA key must be >= 1024 bits for OAEP-SHA256")
        e = 65537
        while True:
            p = _gen_prime(bits // 2)
            q = _gen_prime(bits // 2)
            if p == q:
                continue
            n = p * q
            phi = (p - 1) * (q - 1)
            if phi % e == 0:
                continue
            d = pow(e, -1, phi)
            return cls(n, e, d)

    @property
    def size_bytes(self) -> int:
        return (self.n.bit_length() + 7) // 8

    def public_json(self) -> str:
        return json.dumps({"kty": "RSA", "n": hex(self.n), "e": self.e})

    def private_json(self) -> str:
        return json.dumps({"kty": "RSA", "n": hex(self.n), "e": self.e, "d": hex(self.d)})

    @classmethod
    def from_json(cls, blob: str) -> "RSAKey":
        # junk keys (e.g. a corrupted org public_key record) must fail as
        # ValueError — that's what every crypto call site catches
        try:
            d = json.loads(blob)
            return cls(int(d["n"], 16), int(d["e"]),
                       int(d["d"], 16) if "d" in d else None)
        except ValueError:
            raise
        except Exception as exc:
            raise ValueError("malformed RSA key") from exc

    def public_base64s(self) -> str:
        return bytes_to_base64s(self.public_json().encode())

    @classmethod
    def from_public_base64s(cls, data: str) -> "RSAKey":
        return cls.from_json(base64s_to_bytes(data).decode())

    # -- OAEP --------------------------------------------------------------
    @staticmethod
    def _mgf1(seed: bytes, length: int) -> bytes:
        out = b""
        for c in range((length + _HLEN - 1) // _HLEN):
            out += _HASH(seed + c.to_bytes(4, "big")).digest()
        return out[:length]

    def encrypt(self, msg: bytes) -> bytes:
        k = self.size_bytes
        max_len = k - 2 * _HLEN - 2
        if len(msg) > max_len:
            raise ValueError(f"message too long for RSA-OAEP ({len(msg)} > {max_len})")
        lhash = _HASH(b"").digest()
        ps = b"\x00" * (k - len(msg) - 2 * _HLEN - 2)
        db = lhash + ps + b"\x01" + msg
        seed = secrets.token_bytes(_HLEN)
        masked_db = bytes(a ^ b for a, b in zip(db, self._mgf1(seed, k - _HLEN - 1)))
        masked_seed = bytes(a ^ b for a, b in zip(seed, self._mgf1(masked_db, _HLEN)))
        em = b"\x00" + masked_seed + masked_db
        c = pow(int.from_bytes(em, "big"), self.e, self.n)
        return c.to_bytes(k, "big")

    def decrypt(self, ct: bytes) -> bytes:
        if self.d is None:
            raise ValueError("no private exponent")
        k = self.size_bytes
        m = pow(int.from_bytes(ct, "big"), self.d, self.n)
        em = m.to_bytes(k, "big")
        masked_seed, masked_db = em[1 : 1 + _HLEN], em[1 + _HLEN :]
        seed = bytes(a ^ b for a, b in zip(masked_seed, self._mgf1(masked_db, _HLEN)))
        db = bytes(a ^ b for a, b in zip(masked_db, self._mgf1(seed, k - _HLEN - 1)))
        lhash = _HASH(b"").digest()
        if em[0] != 0 or db[:_HLEN] != lhash:
            raise ValueError("OAEP decryption failed")
        idx = db.find(b"\x01", _HLEN)
        if idx < 0:
            raise ValueError("OAEP decryption failed")
        return db[idx + 1 :]


class RSACryptor(CryptorBase):
    """Hybrid RSA + symmetric cryptor.

    Mirrors ``vantage6.common.encryption.RSACryptor``: construct with the
    path of the organization's private key (created if absent),
    ``encrypt_bytes_to_str(data, recipient_pubkey_b64)`` produces the
    ``$``-joined envelope, ``decrypt_str_to_bytes`` opens one addressed to us.
    """

    def __init__(self, private_key_file: str | Path, bits: int = 2048):
        self.private_key_file = Path(private_key_file)
        if self.private_key_file.exists():
            self.key = RSAKey.from_json(self.private_key_file.read_text())
        else:
            self.key = RSAKey.generate(bits)
            self.private_key_file.parent.mkdir(parents=True, exist_ok=True)
            self.private_key_file.write_text(self.key.private_json())
            os.chmod(self.private_key_file, 0o600)

    @classmethod
    def create_new_rsa_key(cls, path: Path, bits: int = 2048) -> "RSACryptor":
        return cls(path, bits=bits)

    @property
    def public_key_str(self) -> str:
        return self.key.public_base64s()

    def verify_public_key(self, pubkey_base64s: str) -> bool:
        other = RSAKey.from_public_base64s(pubkey_base64s)
        return other.n == self.key.n and other.e == self.key.e

    def encrypt_bytes_to_str(self, data: bytes, pubkey_base64s: str) -> str:
        pub = RSAKey.from_public_base64s(pubkey_base64s)
        shared = secrets.token_bytes(32)
        enc_key = pub.encrypt(shared)
        nonce, ct, tag = _sym_encrypt(shared, data)
        return SEP.join(bytes_to_base64s(x) for x in (enc_key, nonce, ct + tag))

    def decrypt_str_to_bytes(self, data: str) -> bytes:
        fields = [base64s_to_bytes(x) for x in data.split(SEP)]
        if len(fields) == 3:  # reference-shaped: tag folded into field 3
            enc_key, nonce, body = fields
            if len(body) < _HLEN:
                raise ValueError("malformed encryption envelope")
            ct, tag = body[:-_HLEN], body[-_HLEN:]
        elif len(fields) == 4:  # legacy round-1 envelope
            enc_key, nonce, ct, tag = fields
        else:
            raise ValueError("malformed encryption envelope")
        shared = self.key.decrypt(enc_key)
        return _sym_decrypt(shared, nonce, ct, tag)
