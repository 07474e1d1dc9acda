"""Property-based tests (hypothesis) for the codec-critical paths: the
synthetic tokenizer's id round-trip (KV-prefix validity depends on it) and
the embedding blob codec (on-disk format, reference embeddings.ts:116-122)."""
import math

from hypothesis import given, settings
from hypothesis import strategies as st

from room_amd.db.queries import blob_to_vector, vector_to_blob
from room_amd.engine import tokenizer as tok


@settings(max_examples=200, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=tok.VOCAB_SIZE - 1),
                min_size=0, max_size=64))
def test_token_ids_survive_decode_reencode(ids):
    """Generated ids must round-trip through decode→encode exactly — the
    engine reuses KV prefixes by comparing re-encoded session text, so any
    drift would silently invalidate cached prefixes."""
    text = tok.decode(ids)
    again = tok.encode(text)
    assert again == ids


@settings(max_examples=200, deadline=None)
@given(st.text(alphabet=st.characters(blacklist_categories=("Cs",)),
               max_size=200))
def test_encode_always_in_vocab(text):
    ids = tok.encode(text)
    assert all(0 <= i < tok.VOCAB_SIZE for i in ids)


@settings(max_examples=200, deadline=None)
@given(st.lists(st.floats(min_value=-1e6, max_value=1e6, allow_nan=False,
                          width=32), min_size=0, max_size=384))
def test_vector_blob_roundtrip(vec):
    out = blob_to_vector(vector_to_blob(vec))
    assert len(out) == len(vec)
    for a, b in zip(out, vec):
        assert math.isclose(a, b, rel_tol=1e-6, abs_tol=1e-6)


@settings(max_examples=100, deadline=None)
@given(st.lists(st.sampled_from(
    ["system", "user", "assistant"]), min_size=1, max_size=6).flatmap(
        lambda roles: st.tuples(st.just(roles), st.lists(
            st.text(max_size=40), min_size=len(roles), max_size=len(roles)))))
def test_chat_encoding_monotone(roles_contents):
    """encode_chat output grows with the transcript and stays in-vocab."""
    roles, contents = roles_contents
    msgs = [{"role": r, "content": c} for r, c in zip(roles, contents)]
    ids = tok.encode_chat(msgs)
    assert all(0 <= i < tok.VOCAB_SIZE for i in ids)
    longer = tok.encode_chat(msgs + [{"role": "user", "content": "more text"}])
    assert len(longer) > len(ids)
