"""On-chain identity (reference: src/shared/identity.ts — ERC-8004 registration
on Base with data:-URI metadata built from room/queen/workers).

The metadata build, address derivation and registry targets are preserved;
actual broadcast needs chain RPC (absent offline — recorded as pending, the
reference also fail-soft logs RPC errors).
"""
from __future__ import annotations

import base64
import json
import sqlite3

from ..db import queries as q

ERC8004_IDENTITY_REGISTRY = {
    "base": "0x8004A169FB4a3325136EB29fA0ceB6D2e539a432",
    "base-sepolia": "0x8004A818BFB912233c491871b3d84c89A494BD9e",
}
ERC8004_REPUTATION_REGISTRY = {
    "base": "0x8004BAa17C55a88189AE136b182e5fdA19dE9b63",
    "base-sepolia": "0x8004B663056A597Dffe9eCcC1965A193B7388713",
}


def build_agent_metadata(db: sqlite3.Connection, room_id: int) -> dict:
    room = q.get_room(db, room_id)
    if room is None:
        raise ValueError(f"Room {room_id} not found")
    workers = q.list_room_workers(db, room_id)
    queen = next((w for w in workers if w["id"] == room["queen_worker_id"]), None)
    return {
        "name": room.get("queen_nickname") or room["name"],
        "description": room.get("goal") or "",
        "type": "autonomous-agent-room",
        "queen": queen["name"] if queen else None,
        "workers": [{"name": w["name"], "role": w["role"]} for w in workers],
    }


def metadata_data_uri(meta: dict) -> str:
    blob = base64.b64encode(json.dumps(meta).encode()).decode()
    return f"data:application/json;base64,{blob}"


def register_calldata(agent_uri: str) -> bytes:
    """ABI-encoded `register(string agentURI)` call: selector + dynamic
    string head/tail (identity.ts:19-75 contract interface)."""
    from ..utils.crypto import keccak256
    selector = keccak256(b"register(string)")[:4]
    raw = agent_uri.encode()
    padded = raw + b"\x00" * (-len(raw) % 32)
    return (selector + (32).to_bytes(32, "big")
            + len(raw).to_bytes(32, "big") + padded)


def register_identity(db: sqlite3.Connection, room_id: int,
                      chain: str = "base", nonce: int = 0) -> dict:
    """Prepare + SIGN the ERC-8004 registration transaction. The raw tx is
    fully signed offline (RFC6979 ECDSA over the EIP-1559 payload) and ready
    for eth_sendRawTransaction whenever RPC exists; the reference fail-soft
    logs RPC errors the same way."""
    if chain not in ERC8004_IDENTITY_REGISTRY:
        raise ValueError(f"unsupported chain for identity: {chain}")
    wallet = q.get_room_wallet(db, room_id)
    if wallet is None:
        raise ValueError(f"room {room_id} has no wallet")
    meta = build_agent_metadata(db, room_id)
    uri = metadata_data_uri(meta)
    from .constants import CHAIN_CONFIGS
    from .wallet import decrypt_private_key
    raw_tx = None
    try:
        priv = bytes.fromhex(decrypt_private_key(wallet))
        from ..utils.crypto import sign_eip1559_tx
        raw_tx = sign_eip1559_tx(
            priv, CHAIN_CONFIGS[chain]["chainId"], nonce,
            max_priority_fee=10**6, max_fee=10**8, gas=300_000,
            to=ERC8004_IDENTITY_REGISTRY[chain], value=0,
            data=register_calldata(uri))
    except Exception:
        pass          # unsigned prepare still recorded (key unavailable)
    agent_id = f"pending:{chain}:{wallet['address']}"
    q.set_wallet_identity(db, wallet["id"], agent_id)
    q.log_room_activity(db, room_id, "identity",
                        f"Identity registration prepared on {chain}")
    return {"registry": ERC8004_IDENTITY_REGISTRY[chain],
            "address": wallet["address"], "agent_uri": uri,
            "raw_tx": raw_tx,
            "status": ("signed (broadcast when RPC reachable)" if raw_tx
                       else "prepared (wallet key unavailable)")}
