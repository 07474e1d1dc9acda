"""Engine-core tests against real in-memory SQLite (mirrors the reference's
test pyramid base: src/shared/__tests__/ run everything on real SQL)."""
import asyncio
import json

import pytest

from room_amd.core import goals, quorum, room, self_mod, skills
from room_amd.core.agent_loop import AgentLoopManager
from room_amd.core.rate_limit import detect_rate_limit
from room_amd.db import LockedDb
from room_amd.db import queries as q
from room_amd.engine.providers import StubEngine, register_engine


# ------------------------------------------------------------------ room


def test_create_room_creates_queen_root_goal_wallet(db):
    r = room.create_room(db, "alpha", goal="Ship the product", worker_model="stub")
    assert r["queen_worker_id"] is not None
    queen = q.get_worker(db, r["queen_worker_id"])
    assert queen["role"] == "queen"
    assert queen["room_id"] == r["id"]
    gs = q.list_room_goals(db, r["id"])
    assert len(gs) == 1 and gs[0]["description"] == "Ship the product"
    w = q.get_room_wallet(db, r["id"])
    assert w["address"].startswith("0x") and len(w["address"]) == 42
    assert ":" in w["private_key_encrypted"]  # iv:tag:ct
    assert r["webhook_token"]


def test_room_restart_clears_state(db):
    r = room.create_room(db, "beta", goal="objective", worker_model="stub")
    quorum.announce(db, r["id"], r["queen_worker_id"], "prop", "strategy")
    q.create_escalation(db, r["id"], "help?")
    room.restart_room(db, r["id"])
    assert q.list_room_decisions(db, r["id"]) == []
    assert q.list_escalations(db, r["id"]) == []
    gs = q.list_room_goals(db, r["id"])
    assert len(gs) == 1  # root goal recreated


def test_room_status_aggregate(db):
    r = room.create_room(db, "gamma", goal="g", worker_model="stub")
    status = room.get_room_status(db, r["id"])
    assert status["room"]["id"] == r["id"]
    assert len(status["workers"]) == 1
    assert status["token_usage"]["cycles"] == 0


# ------------------------------------------------------------------ quorum


def test_announce_auto_approves_low_impact(db):
    r = room.create_room(db, "ql", worker_model="stub")
    d = quorum.announce(db, r["id"], r["queen_worker_id"], "tiny change", "low_impact")
    assert d["status"] == "approved"


def test_announce_object_flow(db):
    r = room.create_room(db, "qo", worker_model="stub")
    w = q.create_worker(db, "w1", "p", room_id=r["id"])
    d = quorum.announce(db, r["id"], r["queen_worker_id"], "big change", "strategy")
    assert d["status"] == "announced"
    d2 = quorum.object_to(db, d["id"], w["id"], "too risky")
    assert d2["status"] == "objected"
    with pytest.raises(ValueError):
        quorum.object_to(db, d["id"], w["id"], "again")


def test_announce_auto_effective_after_delay(db):
    r = room.create_room(db, "qe", worker_model="stub")
    d = quorum.announce(db, r["id"], r["queen_worker_id"], "change", "strategy",
                        delay_minutes=0)
    assert d["status"] == "announced"
    n = quorum.check_expired_decisions(db)
    assert n == 1
    assert q.get_decision(db, d["id"])["status"] == "effective"


def test_keeper_vote_overrides_announcement(db):
    r = room.create_room(db, "qk", worker_model="stub")
    d = quorum.announce(db, r["id"], r["queen_worker_id"], "x", "strategy")
    out = quorum.keeper_vote(db, d["id"], "no")
    assert out["status"] == "objected"
    d2 = quorum.announce(db, r["id"], r["queen_worker_id"], "y", "strategy")
    out2 = quorum.keeper_vote(db, d2["id"], "yes")
    assert out2["status"] == "effective"


def test_legacy_vote_majority(db):
    r = room.create_room(db, "qv", worker_model="stub")
    w1 = q.create_worker(db, "w1", "p", room_id=r["id"])
    w2 = q.create_worker(db, "w2", "p", room_id=r["id"])
    d = q.create_decision(db, r["id"], r["queen_worker_id"], "vote on it", "strategy")
    quorum.vote(db, d["id"], w1["id"], "yes")
    quorum.vote(db, d["id"], w2["id"], "yes")
    resolved = quorum.resolve_voting_decision(db, d["id"])
    assert resolved["status"] == "approved"
    t = quorum.tally(db, d["id"])
    assert t["yes"] == 2 and t["total"] == 2


# ------------------------------------------------------------------ goals


def test_goal_decompose_and_progress_rollup(db):
    r = room.create_room(db, "gg", goal="root", worker_model="stub")
    root = q.list_room_goals(db, r["id"])[0]
    subs = goals.decompose_goal(db, root["id"], ["a", "b"])
    assert len(subs) == 2
    goals.complete_goal(db, subs[0]["id"])
    parent = q.get_goal(db, root["id"])
    assert parent["progress"] == pytest.approx(0.5)
    goals.update_goal_progress(db, subs[1]["id"], 0.5)
    assert q.get_goal(db, root["id"])["progress"] == pytest.approx(0.75)
    tree = goals.get_goal_tree(db, r["id"])
    assert len(tree) == 1 and len(tree[0]["children"]) == 2


# ------------------------------------------------------------------ skills


def test_skill_activation_and_budget(db):
    r = room.create_room(db, "sk", worker_model="stub")
    skills.create_agent_skill(db, r["id"], "deploy", "how to deploy",
                              activation_context="deploy, release")
    skills.create_agent_skill(db, r["id"], "always", "always on", auto_activate=True)
    block, used = skills.load_skills_for_agent(db, r["id"], "time to DEPLOY now")
    assert "deploy" in block and "always" in block
    assert len(used) == 2
    block2, used2 = skills.load_skills_for_agent(db, r["id"], "nothing relevant")
    assert [s["name"] for s in used2] == ["always"]


def test_skill_budget_caps(db):
    r = room.create_room(db, "sb", worker_model="stub")
    for i in range(12):
        skills.create_agent_skill(db, r["id"], f"s{i}", "x" * 900, auto_activate=True)
    block, used = skills.load_skills_for_agent(db, r["id"], "anything")
    assert len(used) <= 8
    assert len(block) <= 6100


# ------------------------------------------------------------------ self-mod


def test_self_mod_audit_and_revert(db):
    r = room.create_room(db, "sm", worker_model="stub")
    s = skills.create_agent_skill(db, r["id"], "recipe", "v1 content")
    res = self_mod.perform_skill_modification(
        db, r["id"], r["queen_worker_id"], s["id"], "v2 content", reason="improve")
    assert q.get_skill(db, s["id"])["content"] == "v2 content"
    out = self_mod.revert_modification(db, res["audit_id"])
    assert out["reverted"]
    assert q.get_skill(db, s["id"])["content"] == "v1 content"


def test_self_mod_rate_limit_and_forbidden(db):
    r = room.create_room(db, "sm2", worker_model="stub")
    wid = r["queen_worker_id"]
    ok, why = self_mod.can_modify(db, wid, "/home/user/.ssh/id_rsa")
    assert not ok
    s = skills.create_agent_skill(db, r["id"], "x", "c1")
    self_mod.perform_skill_modification(db, r["id"], wid, s["id"], "c2")
    with pytest.raises(PermissionError):
        self_mod.perform_skill_modification(db, r["id"], wid, s["id"], "c3")


# ------------------------------------------------------------------ memory


def test_memory_hybrid_search(db):
    r = room.create_room(db, "mem", worker_model="stub")
    e1 = q.create_entity(db, "pricing research", room_id=r["id"],
                         observations=["competitor charges $99/mo"])
    q.create_entity(db, "deploy runbook", room_id=r["id"],
                    observations=["use blue-green deploys"])
    # keyword-only (no vectors)
    hits = q.hybrid_search(db, "pricing competitor", None, limit=5, room_id=r["id"])
    assert hits and hits[0]["id"] == e1["id"]
    assert "competitor charges $99/mo" in hits[0]["observations"]


def test_memory_semantic_side(db):
    r = room.create_room(db, "mem2", worker_model="stub")
    e1 = q.create_entity(db, "alpha", room_id=r["id"], observations=["a"])
    e2 = q.create_entity(db, "beta", room_id=r["id"], observations=["b"])
    q.upsert_embedding(db, e1["id"], [1.0, 0.0, 0.0], "h1")
    q.upsert_embedding(db, e2["id"], [0.0, 1.0, 0.0], "h2")
    hits = q.hybrid_search(db, "zzz-no-keyword-match", [0.9, 0.1, 0.0],
                           limit=2, room_id=r["id"])
    assert hits[0]["id"] == e1["id"]


def test_embedding_blob_roundtrip(db):
    vec = [0.25, -1.5, 3.0] * 128  # 384 dims
    blob = q.vector_to_blob(vec)
    assert len(blob) == 384 * 4
    back = q.blob_to_vector(blob)
    assert back == pytest.approx(vec)


# ------------------------------------------------------------------ rate limit


def test_rate_limit_detection():
    info = detect_rate_limit("Error 429: Too many requests, retry in 5 minutes")
    assert info.detected
    assert info.wait_ms == 5 * 60 * 1000
    assert not detect_rate_limit("all good").detected
    # clamped to 30s floor / 60min ceiling
    assert detect_rate_limit("rate limit, retry in 2 sec").wait_ms == 30_000
    assert detect_rate_limit("rate limit, in 999 minutes").wait_ms == 3_600_000


# ------------------------------------------------------------------ agent cycle


def test_queen_cycle_decomposes_objective(ldb):
    """BASELINE config 1: one Queen-only goal-decomposition cycle on CPU."""
    with ldb as db:
        r = room.create_room(db, "c1", goal="Build a SaaS", worker_model="stub")
    mgr = AgentLoopManager(ldb)

    async def go():
        return await mgr.run_cycle(r["id"], r["queen_worker_id"])

    out = asyncio.run(go())
    assert out["result"].success
    assert out["result"].tool_calls_executed == 3
    with ldb as db:
        gs = q.list_room_goals(db, r["id"])
        # root + 3 decomposed subgoals
        assert len(gs) == 4
        cycles = q.list_room_cycles(db, r["id"])
        assert cycles[0]["status"] == "completed"
        assert cycles[0]["input_tokens"] > 0
        # queen auto-created an executor worker
        workers = q.list_room_workers(db, r["id"])
        assert any(w["role"] == "executor" for w in workers)
        sess = q.get_agent_session(db, r["id"])


def test_cycle_prompt_contains_context_parts(ldb):
    with ldb as db:
        r = room.create_room(db, "cp", goal="objective X", worker_model="stub")
        q.set_worker_wip(db, r["queen_worker_id"], "was doing step 2")
        q.create_entity(db, "objective X notes", room_id=r["id"],
                        observations=["important fact"])
    mgr = AgentLoopManager(ldb)
    with ldb as db:
        rm = q.get_room(db, r["id"])
        w = q.get_worker(db, r["queen_worker_id"])
        prompt = mgr._build_cycle_prompt(db, rm, w, True, None)
    assert "CONTINUE FORWARD" in prompt
    assert "objective X" in prompt
    assert "Relevant room memory" in prompt
    assert "Workers:" in prompt


def test_agent_loop_trigger_and_stop(ldb):
    with ldb as db:
        r = room.create_room(db, "lp", goal="loop goal", worker_model="stub")
        db.execute("UPDATE rooms SET queen_cycle_gap_ms = 3600000 WHERE id = ?",
                   (r["id"],))
    mgr = AgentLoopManager(ldb)

    async def go():
        state = await mgr.start_agent_loop(r["id"], r["queen_worker_id"])
        # wait for first cycle to complete
        for _ in range(600):
            await asyncio.sleep(0.02)
            if state.cycle_count >= 1:
                break
        assert state.cycle_count >= 1
        # loop is now sleeping 1h; trigger wakes it immediately
        mgr.trigger_agent(r["queen_worker_id"])
        for _ in range(600):
            await asyncio.sleep(0.02)
            if state.cycle_count >= 2:
                break
        assert state.cycle_count >= 2
        mgr.stop_agent(r["queen_worker_id"])
        await asyncio.sleep(0.05)
        assert r["queen_worker_id"] not in mgr.running_loops

    asyncio.run(go())


def test_tool_executor_delegation_wakes_worker(ldb):
    woken = []
    with ldb as db:
        r = room.create_room(db, "dw", goal="obj", worker_model="stub")
        w = q.create_worker(db, "exec1", "p", role="executor", room_id=r["id"])
    mgr = AgentLoopManager(ldb)
    mgr.trigger_agent = lambda wid: woken.append(wid)  # spy
    from room_amd.core import agent_tools
    agent_tools.register_wake_callbacks(mgr.trigger_agent, mgr.wake_room_workers)
    from room_amd.engine.types import ToolCall
    with ldb as db:
        out = agent_tools.execute_agent_tool(
            db, r["id"], r["queen_worker_id"],
            ToolCall("room_delegate_task",
                     {"description": "do it", "worker_id": w["id"]}))
    assert json.loads(out)["delegated_to"] == w["id"]
    assert woken == [w["id"]]


def test_session_persisted_and_rotated(ldb):
    with ldb as db:
        r = room.create_room(db, "sess", goal="obj", worker_model="stub")
    mgr = AgentLoopManager(ldb)

    async def go():
        await mgr.run_cycle(r["id"], r["queen_worker_id"])
        await mgr.run_cycle(r["id"], r["queen_worker_id"])

    asyncio.run(go())
    with ldb as db:
        sess = q.get_agent_session(db, r["queen_worker_id"])
    assert sess is not None and sess["turn_count"] == 2
    msgs = json.loads(sess["messages_json"])
    assert msgs[0]["role"] == "system"


# ------------------------------------------------------------------ wallet/crypto


def test_secret_store_roundtrip():
    from room_amd.core.secret_store import decrypt_secret, encrypt_secret, is_encrypted
    blob = encrypt_secret("api-key-123")
    assert is_encrypted(blob)
    assert decrypt_secret(blob) == "api-key-123"
    assert decrypt_secret("plain") == "plain"


def test_wallet_deterministic_and_decryptable(db):
    r = room.create_room(db, "wl", worker_model="stub")
    w = q.get_room_wallet(db, r["id"])
    from room_amd.core.wallet import decrypt_private_key
    from room_amd.utils.crypto import private_key_to_address
    priv = decrypt_private_key(w)
    assert private_key_to_address(bytes.fromhex(priv)) == w["address"]


def test_send_token_validation(db):
    r = room.create_room(db, "wt", worker_model="stub")
    with pytest.raises(ValueError):
        from room_amd.core.wallet import send_token
        send_token(db, r["id"], "not-an-address", "1.0")
    from room_amd.core.wallet import send_token
    out = send_token(db, r["id"], "0x" + "ab" * 20, "5.0")
    assert out["status"] == "pending"
    w = q.get_room_wallet(db, r["id"])
    txs = q.list_wallet_txs(db, w["id"])
    assert any(t["type"] == "send" and t["amount"] == "5.0" for t in txs)


def test_session_survives_manager_restart(ldb):
    """Session durability: a NEW AgentLoopManager (fresh process semantics)
    resumes the conversation from the agent_sessions row (SURVEY §5
    checkpoint/resume: everything is DB state)."""
    with ldb as db:
        r = room.create_room(db, "restart", goal="persist me", worker_model="stub")
    mgr1 = AgentLoopManager(ldb)
    asyncio.run(mgr1.run_cycle(r["id"], r["queen_worker_id"]))
    with ldb as db:
        sess1 = q.get_agent_session(db, r["queen_worker_id"])
    n_msgs_1 = len(json.loads(sess1["messages_json"]))

    mgr2 = AgentLoopManager(ldb)  # "restarted server"
    asyncio.run(mgr2.run_cycle(r["id"], r["queen_worker_id"]))
    with ldb as db:
        sess2 = q.get_agent_session(db, r["queen_worker_id"])
    assert sess2["turn_count"] == 2
    # the second cycle continued the persisted history
    assert len(json.loads(sess2["messages_json"])) > n_msgs_1


def test_ecdsa_sign_recover_roundtrip():
    from room_amd.utils.crypto import (ecdsa_recover, ecdsa_sign,
                                       generate_private_key, keccak256,
                                       private_key_to_address)
    priv = generate_private_key(b"test-seed")
    addr = private_key_to_address(priv)
    for msg in (b"hello", b"another message", b"\x00" * 32):
        h = keccak256(msg)
        r, s, y = ecdsa_sign(h, priv)
        assert ecdsa_recover(h, r, s, y) == addr


def test_rlp_known_vectors():
    from room_amd.utils.crypto import rlp_encode
    assert rlp_encode(b"dog") == b"\x83dog"
    assert rlp_encode([b"cat", b"dog"]) == b"\xc8\x83cat\x83dog"
    assert rlp_encode(b"") == b"\x80"
    assert rlp_encode(0) == b"\x80"
    assert rlp_encode(15) == b"\x0f"
    assert rlp_encode(1024) == b"\x82\x04\x00"
    assert rlp_encode([]) == b"\xc0"


def test_send_token_produces_signed_raw_tx(db):
    from room_amd.core.wallet import send_token
    r = room.create_room(db, "signtx", worker_model="stub")
    out = send_token(db, r["id"], "0x" + "ab" * 20, "12.5")
    assert out["raw_tx"].startswith("0x02")
    assert len(out["raw_tx"]) > 200


def test_sealed_ballot_min_voters_voter_health(db):
    """Sealed-ballot redaction, min_voters quorum floor, and voter-health
    accounting (reference decisions.ts:97-117, db-queries.ts:1368-1383)."""
    from room_amd.core import quorum, room as room_mod
    from room_amd.db import queries as q

    r = room_mod.create_room(db, "vh", goal="g", worker_model="stub")
    w1 = q.create_worker(db, "a", "p", room_id=r["id"])
    w2 = q.create_worker(db, "b", "p", room_id=r["id"])
    d = q.create_decision(db, r["id"], r["queen_worker_id"], "ship it",
                          "custom", min_voters=2, sealed=True)

    quorum.vote(db, d["id"], w1["id"], "yes")
    # min_voters=2: one vote must NOT resolve
    out = quorum.resolve_voting_decision(db, d["id"])
    assert out["status"] == "voting"

    quorum.vote(db, d["id"], w2["id"], "yes")
    out = quorum.resolve_voting_decision(db, d["id"])
    assert out["status"] == "approved"

    health = q.get_voter_health(db, r["id"])
    by_id = {h["worker_id"]: h for h in health}
    assert by_id[w1["id"]]["votes_cast"] == 1
    assert by_id[w1["id"]]["is_healthy"]


def test_rate_limit_reset_time_formats():
    """All three reset-time formats parse (rate-limit.ts:65-94): clock
    time, unix timestamp, and the relative forms already covered above."""
    import time as _time

    # "resets at H:MM AM/PM" — future clock time within the clamp window
    future = _time.localtime(_time.time() + 10 * 60)
    ampm = "AM" if future.tm_hour < 12 else "PM"
    h12 = future.tm_hour % 12 or 12
    info = detect_rate_limit(
        f"usage limit reached, resets at {h12}:{future.tm_min:02d} {ampm}")
    assert info.detected
    assert 30_000 <= info.wait_ms <= 3_600_000

    # unix timestamp ~7 minutes out
    ts = int(_time.time()) + 420
    info2 = detect_rate_limit(f"rate limited, resets at {ts}")
    assert info2.detected
    assert 6 * 60_000 < info2.wait_ms <= 8 * 60_000

    # detection patterns without any parsable time: default backoff, clamped
    info3 = detect_rate_limit("quota exceeded")
    assert info3.detected and info3.wait_ms >= 30_000
    assert detect_rate_limit("model overloaded, try later").detected
    assert not detect_rate_limit(None).detected


def test_quorum_threshold_modes(db):
    """Room-config vote thresholds (reference README "Quorum Voting"):
    majority (default, queen tie-break), supermajority (≥2/3 of cast
    yes/no), unanimous (any no rejects)."""
    import json as _json

    from room_amd.core import quorum

    def mkroom(threshold):
        r = room.create_room(db, f"th-{threshold}", worker_model="stub")
        cfg = dict(r["config"])
        cfg["threshold"] = threshold
        q.update_room(db, r["id"], config=_json.dumps(cfg))
        ws = [q.create_worker(db, f"w{i}", "p", room_id=r["id"])
              for i in range(3)]
        return q.get_room(db, r["id"]), ws

    def decide(r, ws, votes):
        d = q.create_decision(db, r["id"], r["queen_worker_id"], "p",
                              "high_impact")
        for w, v in zip(ws, votes):
            quorum.vote(db, d["id"], w["id"], v)
        return quorum.resolve_voting_decision(db, d["id"])

    # supermajority: 2/3 yes passes, 2/4... with 3 voters: 2 yes 1 no = 2/3 ✓
    r, ws = mkroom("supermajority")
    assert decide(r, ws, ["yes", "yes", "no"])["status"] == "approved"
    assert decide(r, ws, ["yes", "no", "no"])["status"] == "rejected"
    # simple majority would pass 2-1 either way; supermajority rejects 1-2

    # unanimous: one no rejects even with majority yes
    r2, ws2 = mkroom("unanimous")
    assert decide(r2, ws2, ["yes", "yes", "no"])["status"] == "rejected"
    assert decide(r2, ws2, ["yes", "yes", "yes"])["status"] == "approved"

    # majority (default): 2-1 approves
    r3, ws3 = mkroom("majority")
    assert decide(r3, ws3, ["yes", "yes", "no"])["status"] == "approved"
