from room_amd.engine import tokenizer as tok


def test_encode_deterministic_and_bounded():
    ids = tok.encode("hello world this is a test of the tokenizer")
    assert ids == tok.encode("hello world this is a test of the tokenizer")
    assert all(0 <= i < tok.VOCAB_SIZE for i in ids)
    assert len(ids) >= 9


def test_chat_template_shape():
    msgs = [{"role": "system", "content": "sys"},
            {"role": "user", "content": "hi there"}]
    ids = tok.encode_chat(msgs)
    assert ids.count(tok.IM_START) == 3  # two turns + generation header
    assert ids.count(tok.IM_END) == 2


def test_decode_reencode_roundtrip():
    """Generated ids must survive decode→re-encode so chat-session KV prefixes
    stay valid across turns."""
    gen = [5, 99123, 150000, tok.IM_END]
    text = tok.decode(gen)
    back = tok.encode(text)
    assert back == gen


def test_chat_prefix_extension():
    """Appending a message extends the token stream (prefix preserved)."""
    msgs = [{"role": "system", "content": "s"}, {"role": "user", "content": "a b"}]
    ids1 = tok.encode_chat(msgs)
    msgs2 = msgs + [{"role": "assistant", "content": "⟨7⟩ ⟨8⟩"},
                    {"role": "user", "content": "next"}]
    ids2 = tok.encode_chat(msgs2)
    # common prefix covers everything up to the old generation header
    prefix = ids1[:-2]  # drop trailing <|im_start|> assistant header
    assert ids2[:len(prefix)] == prefix
