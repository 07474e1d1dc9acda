"""Per-room EVM wallet (reference: src/shared/wallet.ts).

Semantics preserved: secp256k1 keygen, AES-256-GCM private-key encryption in
iv:tag:ct hex form, multi-chain token config, tx logging. On-chain RPC calls
are represented but inert in this offline environment (the reference fails
silently on network errors too).
"""
from __future__ import annotations

import os
import sqlite3

from ..db import queries as q
from ..utils.crypto import (decrypt_gcm_hex, encrypt_gcm_hex,
                            generate_private_key, private_key_to_address)
from .constants import CHAIN_CONFIGS, SUPPORTED_CHAINS, SUPPORTED_TOKENS
from .secret_store import machine_key


def create_room_wallet(db: sqlite3.Connection, room_id: int, chain: str = "base",
                       deterministic_seed: str | None = None) -> dict:
    existing = q.get_room_wallet(db, room_id)
    if existing:
        return existing
    priv = generate_private_key(
        deterministic_seed.encode() if deterministic_seed else None)
    address = private_key_to_address(priv)
    encrypted = encrypt_gcm_hex(machine_key(), priv.hex())
    wallet = q.create_wallet_row(db, room_id, address, encrypted, chain)
    q.log_wallet_tx(db, wallet["id"], "fund", "0", description="Wallet created")
    return wallet


def get_wallet_address(db: sqlite3.Connection, room_id: int) -> str | None:
    w = q.get_room_wallet(db, room_id)
    return w["address"] if w else None


def decrypt_private_key(wallet_row: dict) -> str:
    return decrypt_gcm_hex(machine_key(), wallet_row["private_key_encrypted"])


def get_on_chain_balance(db: sqlite3.Connection, room_id: int, chain: str = "base",
                         token: str = "usdc") -> dict:
    """Balance query shape preserved; offline environments report
    unavailable (the reference returns an error dict on RPC failure too)."""
    if chain not in SUPPORTED_CHAINS and chain != "base-sepolia":
        raise ValueError(f"Unsupported chain: {chain}")
    if token not in SUPPORTED_TOKENS:
        raise ValueError(f"Unsupported token: {token}")
    w = q.get_room_wallet(db, room_id)
    if w is None:
        raise ValueError(f"Room {room_id} has no wallet")
    return {
        "address": w["address"], "chain": chain, "token": token,
        "balance": None, "error": "network unavailable",
        "rpcUrl": CHAIN_CONFIGS[chain]["rpcUrl"],
    }


def send_token(db: sqlite3.Connection, room_id: int, to_address: str, amount: str,
               chain: str = "base", token: str = "usdc",
               description: str | None = None) -> dict:
    """Record-keeping half of the transfer path: validates inputs, logs the
    wallet_transactions row with status 'pending' (broadcast requires RPC)."""
    if not to_address.startswith("0x") or len(to_address) != 42:
        raise ValueError("Invalid recipient address")
    try:
        value = float(amount)
    except ValueError as e:
        raise ValueError("Invalid amount") from e
    if value <= 0:
        raise ValueError("Amount must be positive")
    w = q.get_room_wallet(db, room_id)
    if w is None:
        raise ValueError(f"Room {room_id} has no wallet")
    tx_id = q.log_wallet_tx(db, w["id"], "send", amount, counterparty=to_address,
                            description=description, status="pending",
                            category=token)
    q.log_room_activity(db, room_id, "wallet",
                        f"Send {amount} {token.upper()} on {chain} → {to_address}")
    return {"tx_id": tx_id, "status": "pending", "chain": chain, "token": token}
