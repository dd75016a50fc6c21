"""Property-based fuzzing (hypothesis, offline wheel): streaming decode
must equal batch decode for ARBITRARY token-id sequences — byte runs,
incomplete UTF-8, specials, unassigned pseudo-word ids, interleavings.
The incremental decoders carry state across pushes (byte buffers,
holdback windows); equivalence with the stateless batch decode is the
invariant every streaming client depends on."""

from hypothesis import given, settings, strategies as st

from gridllm_amd.engine.tokenizer import ByteTokenizer


@settings(max_examples=300, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=128255), max_size=80))
def test_byte_stream_decode_equals_batch(ids):
    tok = ByteTokenizer(vocab_size=128256)
    dec = tok.decode_stream()
    out = "".join(dec.push(i) for i in ids) + dec.flush()
    assert out == tok.decode(ids)


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=120))
def test_byte_tokenizer_text_roundtrip(text):
    tok = ByteTokenizer()
    assert tok.decode(tok.encode(text, bos=False)) == text


@settings(max_examples=500, deadline=None)
@given(st.lists(st.one_of(
    st.integers(min_value=0, max_value=255),      # byte-heavy: UTF-8 edges
    st.integers(min_value=256, max_value=128255)),
    max_size=60))
def test_byte_stream_decode_equals_batch_bytes_heavy(ids):
    """Byte-dense sequences hit the incremental-UTF-8 edges (split code
    points, invalid leads, holdback across pseudo-word flushes) that the
    uniform-id fuzz almost never generates — this strategy found the
    4-byte-bailout code-point split."""
    tok = ByteTokenizer(vocab_size=128256)
    dec = tok.decode_stream()
    out = "".join(dec.push(i) for i in ids) + dec.flush()
    assert out == tok.decode(ids)


def test_hf_stream_decoder_fuzz_roundtrip(tmp_path_factory):
    """HF span decoder: for any real text, streaming the encoded ids
    reassembles the batch decode exactly (code points split across BPE
    tokens included)."""
    from tokenizers import (Tokenizer, decoders, models, pre_tokenizers,
                            trainers)
    tok = Tokenizer(models.BPE())
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    tr = trainers.BpeTrainer(
        vocab_size=300, special_tokens=["<|end_of_text|>"],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet())
    tok.train_from_iterator(["hello wörld ✓ 日本語 {json: true}"] * 10, tr)
    path = str(tmp_path_factory.mktemp("tokf") / "tokenizer.json")
    tok.save(path)
    from gridllm_amd.engine.tokenizer import HFTokenizer
    hf = HFTokenizer(path)

    from hypothesis import given as _given, settings as _settings
    from hypothesis import strategies as _st

    @_settings(max_examples=150, deadline=None)
    @_given(_st.text(max_size=60))
    def check(text):
        ids = hf.encode(text)
        dec = hf.decode_stream()
        out = "".join(dec.push(i) for i in ids) + dec.flush()
        assert out == hf.decode(ids)

    check()


@settings(max_examples=400, deadline=None)
@given(st.text(alphabet="<think/>abc\n ", max_size=60),
       st.lists(st.integers(min_value=1, max_value=8), max_size=20))
def test_think_splitter_chunking_invariant(text, cuts):
    """ThinkSplitter: any chunking of the same text yields the same
    (thinking, content) split as the batch split_think — the holdback
    window for a partial close tag must not leak or drop characters.
    Alphabet is tag-dense to hit partial-tag boundaries constantly."""
    from gridllm_amd.engine.chat import ThinkSplitter, split_think
    sp = ThinkSplitter()
    th, co = [], []
    pos = 0
    for c in cuts:
        piece, pos = text[pos:pos + c], pos + c
        t, x = sp.push(piece)
        th.append(t)
        co.append(x)
    t, x = sp.push(text[pos:])
    th.append(t)
    co.append(x)
    t, x = sp.flush()
    th.append(t)
    co.append(x)
    # both consume the GENERATION only (the opening tag is part of the
    # prompt; generation starts inside the think block)
    want_th, want_co = split_think(text)
    assert "".join(th) == want_th, (text, cuts)
    assert "".join(co) == want_co, (text, cuts)
