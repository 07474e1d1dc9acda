"""Self-contained crypto primitives for wallet/secret semantics.

The reference relies on Node's crypto (AES-256-GCM, secret-store.ts:1-42,
wallet.ts:56-83) and viem (secp256k1 keygen + keccak address derivation,
wallet.ts:10-12). This image has no Python crypto package, so the primitives
are implemented here directly. They protect small at-rest payloads (private
keys, credentials); throughput is irrelevant.
"""
from __future__ import annotations

import hashlib
import hmac
import os
import struct

# ------------------------------------------------------------------ Keccak-256
# (Ethereum's keccak — original padding 0x01, not NIST SHA-3's 0x06)

_KECCAK_ROUNDS = 24
_RC = [
    0x0000000000000001, 0x0000000000008082, 0x800000000000808A, 0x8000000080008000,
    0x000000000000808B, 0x0000000080000001, 0x8000000080008081, 0x8000000000008009,
    0x000000000000008A, 0x0000000000000088, 0x0000000080008009, 0x000000008000000A,
    0x000000008000808B, 0x800000000000008B, 0x8000000000008089, 0x8000000000008003,
    0x8000000000008002, 0x8000000000000080, 0x000000000000800A, 0x800000008000000A,
    0x8000000080008081, 0x8000000000008080, 0x0000000080000001, 0x8000000080008008,
]
_ROT = [[0, 36, 3, 41, 18], [1, 44, 10, 45, 2], [62, 6, 43, 15, 61],
        [28, 55, 25, 21, 56], [27, 20, 39, 8, 14]]

_M64 = (1 << 64) - 1


def _rol(x: int, n: int) -> int:
    n %= 64
    return ((x << n) | (x >> (64 - n))) & _M64


def _keccak_f(state: list[int]) -> None:
    for rnd in range(_KECCAK_ROUNDS):
        # theta
        c = [state[x] ^ state[x + 5] ^ state[x + 10] ^ state[x + 15] ^ state[x + 20]
             for x in range(5)]
        d = [c[(x - 1) % 5] ^ _rol(c[(x + 1) % 5], 1) for x in range(5)]
        for x in range(5):
            for y in range(5):
                state[x + 5 * y] ^= d[x]
        # rho + pi
        b = [0] * 25
        for x in range(5):
            for y in range(5):
                b[y + 5 * ((2 * x + 3 * y) % 5)] = _rol(state[x + 5 * y], _ROT[x][y])
        # chi
        for x in range(5):
            for y in range(5):
                state[x + 5 * y] = b[x + 5 * y] ^ ((~b[(x + 1) % 5 + 5 * y]) & _M64
                                                   & b[(x + 2) % 5 + 5 * y])
        # iota
        state[0] ^= _RC[rnd]


def keccak256(data: bytes) -> bytes:
    rate = 136  # 1088 bits
    state = [0] * 25
    # pad10*1 with domain byte 0x01 (keccak, not sha3)
    padded = bytearray(data)
    padded.append(0x01)
    while len(padded) % rate:
        padded.append(0x00)
    padded[-1] |= 0x80
    for block_off in range(0, len(padded), rate):
        for i in range(rate // 8):
            state[i] ^= struct.unpack_from("<Q", padded, block_off + i * 8)[0]
        _keccak_f(state)
    return b"".join(struct.pack("<Q", state[i]) for i in range(4))


# ------------------------------------------------------------------ secp256k1

_P = 2**256 - 2**32 - 977
_N = 0xFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFEBAAEDCE6AF48A03BBFD25E8CD0364141
_G = (
    0x79BE667EF9DCBBAC55A06295CE870B07029BFCDB2DCE28D959F2815B16F81798,
    0x483ADA7726A3C4655DA4FBFC0E1108A8FD17B448A68554199C47D08FFB10D4B8,
)


def _inv(a: int, m: int) -> int:
    return pow(a, m - 2, m)


def _ec_add(p1, p2):
    if p1 is None:
        return p2
    if p2 is None:
        return p1
    x1, y1 = p1
    x2, y2 = p2
    if x1 == x2 and (y1 + y2) % _P == 0:
        return None
    if p1 == p2:
        lam = (3 * x1 * x1) * _inv(2 * y1, _P) % _P
    else:
        lam = (y2 - y1) * _inv(x2 - x1, _P) % _P
    x3 = (lam * lam - x1 - x2) % _P
    y3 = (lam * (x1 - x3) - y1) % _P
    return (x3, y3)


def _ec_mul(k: int, point=_G):
    result = None
    addend = point
    while k:
        if k & 1:
            result = _ec_add(result, addend)
        addend = _ec_add(addend, addend)
        k >>= 1
    return result


def generate_private_key(seed: bytes | None = None) -> bytes:
    """32-byte secp256k1 private key; deterministic when a seed is provided
    (room wallets derive from a SHA-256 seed, reference room.ts:52-62)."""
    while True:
        raw = hashlib.sha256(seed).digest() if seed else os.urandom(32)
        k = int.from_bytes(raw, "big")
        if 0 < k < _N:
            return raw
        seed = raw


def private_key_to_address(priv: bytes) -> str:
    """EVM address = last 20 bytes of keccak256(uncompressed pubkey minus 0x04)."""
    k = int.from_bytes(priv, "big")
    pub = _ec_mul(k)
    assert pub is not None
    raw = pub[0].to_bytes(32, "big") + pub[1].to_bytes(32, "big")
    addr = keccak256(raw)[-20:]
    return to_checksum_address("0x" + addr.hex())


def to_checksum_address(addr: str) -> str:
    """EIP-55 checksum casing."""
    body = addr.lower().replace("0x", "")
    digest = keccak256(body.encode()).hex()
    out = "0x" + "".join(
        c.upper() if int(digest[i], 16) >= 8 else c for i, c in enumerate(body))
    return out


# ------------------------------------------------------------------ AES-256-GCM

_SBOX = None
_INV_SBOX = None


def _build_sbox():
    global _SBOX, _INV_SBOX
    if _SBOX is not None:
        return
    sbox = [0] * 256
    p = q_ = 1
    sbox[0] = 0x63
    while True:
        # multiply p by 3
        p = p ^ ((p << 1) & 0xFF) ^ (0x1B if p & 0x80 else 0)
        # divide q by 3
        q_ ^= (q_ << 1) & 0xFF
        q_ ^= (q_ << 2) & 0xFF
        q_ ^= (q_ << 4) & 0xFF
        if q_ & 0x80:
            q_ ^= 0x09
        xformed = (q_ ^ ((q_ << 1) | (q_ >> 7)) ^ ((q_ << 2) | (q_ >> 6))
                   ^ ((q_ << 3) | (q_ >> 5)) ^ ((q_ << 4) | (q_ >> 4))) & 0xFF
        sbox[p] = xformed ^ 0x63
        if p == 1:
            break
    inv = [0] * 256
    for i, v in enumerate(sbox):
        inv[v] = i
    _SBOX, _INV_SBOX = sbox, inv


def _xtime(a: int) -> int:
    a <<= 1
    if a & 0x100:
        a ^= 0x11B
    return a & 0xFF


def _mul(a: int, b: int) -> int:
    r = 0
    while b:
        if b & 1:
            r ^= a
        a = _xtime(a)
        b >>= 1
    return r


def _expand_key(key: bytes) -> list[list[int]]:
    _build_sbox()
    nk = len(key) // 4  # 8 for AES-256
    nr = nk + 6
    w = [list(key[4 * i:4 * i + 4]) for i in range(nk)]
    rcon = 1
    for i in range(nk, 4 * (nr + 1)):
        temp = list(w[i - 1])
        if i % nk == 0:
            temp = temp[1:] + temp[:1]
            temp = [_SBOX[b] for b in temp]
            temp[0] ^= rcon
            rcon = _xtime(rcon)
        elif nk > 6 and i % nk == 4:
            temp = [_SBOX[b] for b in temp]
        w.append([w[i - nk][j] ^ temp[j] for j in range(4)])
    return [sum(w[4 * r:4 * r + 4], []) for r in range(nr + 1)]


def _aes_encrypt_block(block: bytes, round_keys: list[list[int]]) -> bytes:
    _build_sbox()
    state = [block[i] ^ round_keys[0][i] for i in range(16)]
    nr = len(round_keys) - 1
    for rnd in range(1, nr + 1):
        state = [_SBOX[b] for b in state]
        # shift rows (column-major state layout)
        s = state
        state = [
            s[0], s[5], s[10], s[15],
            s[4], s[9], s[14], s[3],
            s[8], s[13], s[2], s[7],
            s[12], s[1], s[6], s[11],
        ]
        if rnd != nr:
            mixed = []
            for c in range(4):
                col = state[4 * c:4 * c + 4]
                mixed += [
                    _mul(col[0], 2) ^ _mul(col[1], 3) ^ col[2] ^ col[3],
                    col[0] ^ _mul(col[1], 2) ^ _mul(col[2], 3) ^ col[3],
                    col[0] ^ col[1] ^ _mul(col[2], 2) ^ _mul(col[3], 3),
                    _mul(col[0], 3) ^ col[1] ^ col[2] ^ _mul(col[3], 2),
                ]
            state = mixed
        state = [state[i] ^ round_keys[rnd][i] for i in range(16)]
    return bytes(state)


def _ghash_mul(x: int, y: int) -> int:
    # GF(2^128) multiply, reflected per GCM spec
    r = 0xE1000000000000000000000000000000
    z = 0
    v = x
    for i in range(128):
        if (y >> (127 - i)) & 1:
            z ^= v
        if v & 1:
            v = (v >> 1) ^ r
        else:
            v >>= 1
    return z


class AESGCM:
    def __init__(self, key: bytes):
        if len(key) not in (16, 24, 32):
            raise ValueError("key must be 128/192/256-bit")
        self._rk = _expand_key(key)
        self._h = int.from_bytes(_aes_encrypt_block(b"\x00" * 16, self._rk), "big")

    def _ctr(self, j0: bytes, data: bytes) -> bytes:
        out = bytearray()
        counter = int.from_bytes(j0, "big")
        for off in range(0, len(data), 16):
            counter = (counter & ~0xFFFFFFFF) | ((counter + 1) & 0xFFFFFFFF)
            ks = _aes_encrypt_block(counter.to_bytes(16, "big"), self._rk)
            chunk = data[off:off + 16]
            out += bytes(a ^ b for a, b in zip(chunk, ks))
        return bytes(out)

    def _ghash(self, aad: bytes, ct: bytes) -> bytes:
        def pad(b: bytes) -> bytes:
            return b + b"\x00" * ((16 - len(b) % 16) % 16)

        blob = pad(aad) + pad(ct) + struct.pack(">QQ", len(aad) * 8, len(ct) * 8)
        y = 0
        for off in range(0, len(blob), 16):
            y = _ghash_mul(y ^ int.from_bytes(blob[off:off + 16], "big"), self._h)
        return y.to_bytes(16, "big")

    def encrypt(self, iv: bytes, plaintext: bytes, aad: bytes = b"") -> tuple[bytes, bytes]:
        """Returns (ciphertext, 16-byte tag)."""
        if len(iv) != 12:
            raise ValueError("GCM iv must be 12 bytes")
        j0 = iv + b"\x00\x00\x00\x01"
        ct = self._ctr(j0, plaintext)
        s = self._ghash(aad, ct)
        ek_j0 = _aes_encrypt_block(j0, self._rk)
        tag = bytes(a ^ b for a, b in zip(ek_j0, s))
        return ct, tag

    def decrypt(self, iv: bytes, ciphertext: bytes, tag: bytes, aad: bytes = b"") -> bytes:
        if len(iv) != 12:
            raise ValueError("GCM iv must be 12 bytes")
        j0 = iv + b"\x00\x00\x00\x01"
        s = self._ghash(aad, ciphertext)
        ek_j0 = _aes_encrypt_block(j0, self._rk)
        expect = bytes(a ^ b for a, b in zip(ek_j0, s))
        if not hmac.compare_digest(expect, tag):
            raise ValueError("GCM tag mismatch")
        return self._ctr(j0, ciphertext)


def encrypt_gcm_hex(key: bytes, plaintext: str) -> str:
    """iv:tag:ciphertext hex format (reference wallet.ts:56-83)."""
    iv = os.urandom(12)
    ct, tag = AESGCM(key).encrypt(iv, plaintext.encode())
    return f"{iv.hex()}:{tag.hex()}:{ct.hex()}"


def decrypt_gcm_hex(key: bytes, blob: str) -> str:
    iv_hex, tag_hex, ct_hex = blob.split(":")
    pt = AESGCM(key).decrypt(bytes.fromhex(iv_hex), bytes.fromhex(ct_hex),
                             bytes.fromhex(tag_hex))
    return pt.decode()


# ------------------------------------------------------------------ RLP

def rlp_encode(item) -> bytes:
    """Recursive-length-prefix encoding (Ethereum wire format)."""
    if isinstance(item, int):
        if item == 0:
            item = b""
        else:
            item = item.to_bytes((item.bit_length() + 7) // 8, "big")
    if isinstance(item, str):
        item = bytes.fromhex(item[2:] if item.startswith("0x") else item)
    if isinstance(item, (bytes, bytearray)):
        if len(item) == 1 and item[0] < 0x80:
            return bytes(item)
        return _rlp_len(len(item), 0x80) + bytes(item)
    if isinstance(item, (list, tuple)):
        payload = b"".join(rlp_encode(x) for x in item)
        return _rlp_len(len(payload), 0xC0) + payload
    raise TypeError(f"cannot RLP-encode {type(item)}")


def _rlp_len(n: int, offset: int) -> bytes:
    if n < 56:
        return bytes([offset + n])
    blen = n.to_bytes((n.bit_length() + 7) // 8, "big")
    return bytes([offset + 55 + len(blen)]) + blen


# ---------------------------------------------------------- ECDSA (secp256k1)

def _rfc6979_k(msg_hash: bytes, priv: bytes) -> int:
    """Deterministic nonce per RFC 6979 (HMAC-SHA256)."""
    V = b"\x01" * 32
    K = b"\x00" * 32
    K = hmac.new(K, V + b"\x00" + priv + msg_hash, hashlib.sha256).digest()
    V = hmac.new(K, V, hashlib.sha256).digest()
    K = hmac.new(K, V + b"\x01" + priv + msg_hash, hashlib.sha256).digest()
    V = hmac.new(K, V, hashlib.sha256).digest()
    while True:
        V = hmac.new(K, V, hashlib.sha256).digest()
        k = int.from_bytes(V, "big")
        if 0 < k < _N:
            return k
        K = hmac.new(K, V + b"\x00", hashlib.sha256).digest()
        V = hmac.new(K, V, hashlib.sha256).digest()


def ecdsa_sign(msg_hash: bytes, priv: bytes) -> tuple[int, int, int]:
    """Returns (r, s, y_parity) with low-s normalization (EIP-2)."""
    z = int.from_bytes(msg_hash, "big")
    d = int.from_bytes(priv, "big")
    k = _rfc6979_k(msg_hash, priv)
    R = _ec_mul(k)
    r = R[0] % _N
    s = (_inv(k, _N) * (z + r * d)) % _N
    y_parity = R[1] & 1
    if s > _N // 2:
        s = _N - s
        y_parity ^= 1
    return r, s, y_parity


def ecdsa_recover(msg_hash: bytes, r: int, s: int, y_parity: int) -> str:
    """Recovers the signer's EVM address (used as a self-check in tests)."""
    z = int.from_bytes(msg_hash, "big")
    x = r
    # y² = x³ + 7 mod p
    y = pow((x * x * x + 7) % _P, (_P + 1) // 4, _P)
    if y & 1 != y_parity:
        y = _P - y
    Rpt = (x, y)
    r_inv = _inv(r, _N)
    # Q = r⁻¹ (s·R − z·G)
    sR = _ec_mul(s, Rpt)
    zG = _ec_mul(z % _N)
    neg_zG = (zG[0], _P - zG[1])
    Q = _ec_mul(r_inv, _ec_add(sR, neg_zG))
    raw = Q[0].to_bytes(32, "big") + Q[1].to_bytes(32, "big")
    return to_checksum_address("0x" + keccak256(raw)[-20:].hex())


# ------------------------------------------------------------ EIP-1559 tx

def erc20_transfer_calldata(to_address: str, amount: int) -> bytes:
    """transfer(address,uint256) calldata."""
    selector = keccak256(b"transfer(address,uint256)")[:4]
    addr = bytes.fromhex(to_address[2:].lower()).rjust(32, b"\x00")
    return selector + addr + amount.to_bytes(32, "big")


def sign_eip1559_tx(priv: bytes, chain_id: int, nonce: int,
                    max_priority_fee: int, max_fee: int, gas: int,
                    to: str, value: int, data: bytes = b"") -> str:
    """Builds and signs a type-2 (EIP-1559) transaction; returns raw tx hex
    ready for eth_sendRawTransaction."""
    fields = [chain_id, nonce, max_priority_fee, max_fee, gas, to, value,
              data, []]  # empty access list
    unsigned = b"\x02" + rlp_encode(fields)
    r, s, y = ecdsa_sign(keccak256(unsigned), priv)
    signed = b"\x02" + rlp_encode(fields + [y, r, s])
    return "0x" + signed.hex()
