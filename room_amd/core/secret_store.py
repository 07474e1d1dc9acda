"""App-level secret encryption with `enc:v1:` prefix and machine-derived key
(reference: src/shared/secret-store.ts — hostname+user key unless overridden
by env)."""
from __future__ import annotations

import getpass
import hashlib
import os
import socket

from ..utils.crypto import decrypt_gcm_hex, encrypt_gcm_hex

PREFIX = "enc:v1:"
ENV_KEY = "ROOMAMD_SECRET_KEY"


def machine_key() -> bytes:
    override = os.environ.get(ENV_KEY) or os.environ.get("QUOROOM_SECRET_KEY")
    if override:
        return hashlib.sha256(override.encode()).digest()
    material = f"{socket.gethostname()}:{getpass.getuser()}"
    return hashlib.sha256(material.encode()).digest()


def encrypt_secret(plaintext: str) -> str:
    return PREFIX + encrypt_gcm_hex(machine_key(), plaintext)


def decrypt_secret(blob: str) -> str:
    if not blob.startswith(PREFIX):
        return blob  # legacy/plain value passes through (reference behavior)
    return decrypt_gcm_hex(machine_key(), blob[len(PREFIX):])


def is_encrypted(blob: str) -> bool:
    return blob.startswith(PREFIX)
