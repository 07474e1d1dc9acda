"""Quorum governance — announce-and-object model (reference: src/shared/quorum.ts).

Queen announces; decision auto-effective after `delay_minutes` (default 10)
unless a worker objects first (quorum.ts:17-48,73-95). Auto-approve list per
decision type from room config (quorum.ts:23-33). Keeper vote override
(quorum.ts:112-133). Legacy vote path preserved (quorum.ts:100-110).

In the multi-GPU swarm, vote aggregation additionally runs as an RCCL
all-gather over xGMI (room_amd.parallel.swarm.quorum_allgather); SQLite rows
remain the durable source of truth.
"""
from __future__ import annotations

import sqlite3

from ..db import queries as q
from .constants import ANNOUNCE_DEFAULT_DELAY_MINUTES


def announce(db: sqlite3.Connection, room_id: int, proposer_id: int | None,
             proposal: str, decision_type: str = "low_impact",
             delay_minutes: int | None = None) -> dict:
    room = q.get_room(db, room_id)
    if room is None:
        raise ValueError(f"Room {room_id} not found")

    if decision_type in room["config"].get("autoApprove", []):
        decision = q.create_decision(db, room_id, proposer_id, proposal,
                                     decision_type, "majority")
        q.resolve_decision(db, decision["id"], "approved", "Auto-approved")
        q.log_room_activity(db, room_id, "decision",
                            f"Auto-approved: {proposal}", actor_id=proposer_id)
        return q.get_decision(db, decision["id"])

    delay = delay_minutes if delay_minutes is not None else ANNOUNCE_DEFAULT_DELAY_MINUTES
    effective_at = q._iso_in_ms(delay * 60_000)
    decision = q.create_announcement(db, room_id, proposer_id, proposal,
                                     decision_type, effective_at)
    q.log_room_activity(db, room_id, "decision",
                        f"Announced: {proposal} (effective in {delay} min)",
                        actor_id=proposer_id)
    return decision


# Back-compat alias used by MCP tools (reference exports announce as propose).
propose = announce


def object_to(db: sqlite3.Connection, decision_id: int, worker_id: int,
              reason: str) -> dict:
    decision = q.get_decision(db, decision_id)
    if decision is None:
        raise ValueError(f"Decision {decision_id} not found")
    if decision["status"] != "announced":
        raise ValueError(
            f"Decision {decision_id} is not open for objection (status: {decision['status']})")
    q.resolve_decision(db, decision_id, "objected",
                       f"Objected by worker #{worker_id}: {reason}")
    q.log_room_activity(db, decision["room_id"], "decision",
                        f"Objected: {decision['proposal']} — {reason}",
                        actor_id=worker_id)
    return q.get_decision(db, decision_id)


def check_expired_decisions(db: sqlite3.Connection) -> int:
    """Resolve announcements past effective_at (auto-effective) and expired
    legacy voting decisions. Called at the top of every agent cycle."""
    count = 0
    for d in q.get_announced_decisions_past_effective(db):
        q.resolve_decision(db, d["id"], "effective", "No objections — auto-effective")
        q.log_room_activity(db, d["room_id"], "decision",
                            f"Effective: {d['proposal']} (no objections)")
        count += 1
    for d in q.get_expired_voting_decisions(db):
        q.resolve_decision(db, d["id"], "expired", "Voting period expired")
        q.log_room_activity(db, d["room_id"], "decision", f"Expired: {d['proposal']}")
        count += 1
    return count


def vote(db: sqlite3.Connection, decision_id: int, worker_id: int, vote_value: str,
         reasoning: str | None = None) -> dict:
    decision = q.get_decision(db, decision_id)
    if decision is None:
        raise ValueError(f"Decision {decision_id} not found")
    if decision["status"] != "voting":
        raise ValueError(
            f"Decision {decision_id} is not open for voting (status: {decision['status']})")
    return q.cast_vote(db, decision_id, worker_id, vote_value, reasoning)


def tally(db: sqlite3.Connection, decision_id: int) -> dict:
    """Vote tally (quorum.ts:100-110 semantics). When a SwarmContext is
    installed (multi-GPU swarm, one room shard per rank), per-shard counts
    are summed with an RCCL all-reduce over xGMI so the tally covers votes
    cast on every GPU's shard — the collective is part of the quorum system,
    not a bench bolt-on. Local SQLite rows remain the durable record."""
    votes = q.get_votes(db, decision_id)
    yes = sum(1 for v in votes if v["vote"] == "yes")
    no = sum(1 for v in votes if v["vote"] == "no")
    abstain = sum(1 for v in votes if v["vote"] == "abstain")
    local = {"yes": yes, "no": no, "abstain": abstain, "total": len(votes)}
    from ..parallel.swarm import get_swarm_context
    ctx = get_swarm_context()
    if ctx is not None and ctx.collective_safe:
        agg = ctx.tally_allreduce(local)
        agg["source"] = f"rccl-allreduce world={ctx.world_size}"
        return agg
    local["source"] = "local"
    return local


def resolve_voting_decision(db: sqlite3.Connection, decision_id: int) -> dict:
    """Majority resolution with queen tie-breaker semantics. min_voters on
    the decision blocks resolution until enough votes are in; non-voters
    are charged a missed vote for voter-health accounting."""
    decision = q.get_decision(db, decision_id)
    if decision is None:
        raise ValueError(f"Decision {decision_id} not found")
    t = tally(db, decision_id)
    if t["total"] < (decision.get("min_voters") or 0):
        return decision  # quorum not met yet — stays open
    voted = {v["worker_id"] for v in q.get_votes(db, decision_id)}
    for w in q.list_room_workers(db, decision["room_id"]):
        if w["id"] not in voted and w["id"] != decision.get("proposer_id"):
            db.execute("UPDATE workers SET votes_missed = votes_missed + 1"
                       " WHERE id = ?", (w["id"],))
    src = t.get("source", "local")
    room = q.get_room(db, decision["room_id"])
    threshold = (room["config"].get("threshold", "majority")
                 if room else "majority")
    yes, no = t["yes"], t["no"]
    if threshold == "unanimous":
        # every cast yes/no vote must be yes (README "unanimous")
        if yes > 0 and no == 0:
            q.resolve_decision(db, decision_id, "approved",
                               f"{yes}-0 unanimous ({src})")
        else:
            q.resolve_decision(db, decision_id, "rejected",
                               f"{yes}-{no} not unanimous ({src})")
    elif threshold == "supermajority":
        # ≥2/3 of cast yes/no votes
        if (yes + no) > 0 and 3 * yes >= 2 * (yes + no):
            q.resolve_decision(db, decision_id, "approved",
                               f"{yes}-{no} supermajority ({src})")
        else:
            q.resolve_decision(db, decision_id, "rejected",
                               f"{yes}-{no} below 2/3 ({src})")
    elif yes > no:
        q.resolve_decision(db, decision_id, "approved",
                           f"{yes}-{no} ({src})")
    elif no > yes:
        q.resolve_decision(db, decision_id, "rejected",
                           f"{yes}-{no} ({src})")
    else:
        tie_breaker = room["config"].get("tieBreaker", "queen") if room else "queen"
        if tie_breaker == "queen" and decision["proposer_id"] == (
                room or {}).get("queen_worker_id"):
            q.resolve_decision(db, decision_id, "approved", "tie — queen tie-breaker")
        else:
            q.resolve_decision(db, decision_id, "rejected", "tie")
    return q.get_decision(db, decision_id)


def keeper_vote(db: sqlite3.Connection, decision_id: int, vote_value: str) -> dict:
    decision = q.get_decision(db, decision_id)
    if decision is None:
        raise ValueError(f"Decision {decision_id} not found")
    if decision["status"] == "announced":
        if vote_value == "no":
            q.resolve_decision(db, decision_id, "objected", "Keeper objected")
        else:
            q.resolve_decision(db, decision_id, "effective", "Keeper approved")
        return q.get_decision(db, decision_id)
    if decision["status"] != "voting":
        raise ValueError(
            f"Decision {decision_id} is not open for voting (status: {decision['status']})")
    q.set_keeper_vote(db, decision_id, vote_value)
    return q.get_decision(db, decision_id)
