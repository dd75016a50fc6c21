"""Grammar-constrained decoding: a byte-level JSON pushdown automaton.

Ollama's `format: "json"` forces syntactically valid JSON output; the
reference forwarded the flag to Ollama (server/src/routes/ollama.ts:45).
Here the engine enforces it directly when the tokenizer is byte-level
(the default self-contained tokenizer): at each step the automaton yields
the set of legal next BYTES and the sampler masks every other token.
EOS becomes legal exactly when a complete top-level value has been
emitted.
"""

from __future__ import annotations

_WS = frozenset(b" \t\n\r")
_DIGITS = frozenset(b"0123456789")
_HEX = frozenset(b"0123456789abcdefABCDEF")
# any byte legal inside a JSON string except '"' and '\' and C0 controls
_STRING_BODY = frozenset(b for b in range(0x20, 0x100) if b not in (0x22, 0x5C))

# parser modes
(VALUE, STRING, STR_ESC, STR_U1, STR_U2, STR_U3, STR_U4, LITERAL,
 INT_NEG, INT, INT_ZERO, FRAC_START, FRAC, EXP_START, EXP_SIGNED, EXP,
 AFTER, OBJ_FIRST, OBJ_KEY_START, OBJ_COLON, ARR_FIRST) = range(21)

_LITERALS = {ord("t"): b"rue", ord("f"): b"alse", ord("n"): b"ull"}


class JsonByteConstraint:
    """allowed() -> (frozenset[int] legal next bytes, eos_ok); push(byte)
    advances the automaton. Bytes not in allowed() must be masked out."""

    def __init__(self) -> None:
        self.stack: list[int] = []      # of ord('{') / ord('[')
        self.mode = VALUE
        self._lit = b""                 # remaining literal suffix
        self._key = False               # current string is an object key

    # ------------------------------------------------------------- allowed

    def allowed(self) -> tuple[frozenset, bool]:
        m = self.mode
        if m == VALUE:
            return (_WS | _DIGITS | frozenset(b'{["-tfn'), False)
        if m == STRING:
            return (_STRING_BODY | {0x22, 0x5C}, False)
        if m == STR_ESC:
            return (frozenset(b'"\\/bfnrtu'), False)
        if m in (STR_U1, STR_U2, STR_U3, STR_U4):
            return (_HEX, False)
        if m == LITERAL:
            return (frozenset({self._lit[0]}), False)
        if m == INT_NEG:
            return (_DIGITS, False)
        if m == INT:
            return (_DIGITS | frozenset(b".eE") | self._close_set(),
                    self._eos_ok())
        if m == INT_ZERO:
            return (frozenset(b".eE") | self._close_set(), self._eos_ok())
        if m == FRAC_START:
            return (_DIGITS, False)
        if m == FRAC:
            return (_DIGITS | frozenset(b"eE") | self._close_set(),
                    self._eos_ok())
        if m == EXP_START:
            return (_DIGITS | frozenset(b"+-"), False)
        if m == EXP_SIGNED:
            return (_DIGITS, False)
        if m == EXP:
            return (_DIGITS | self._close_set(), self._eos_ok())
        if m == AFTER:
            return (self._close_set(), self._eos_ok())
        if m == OBJ_FIRST:
            return (_WS | frozenset(b'"}'), False)
        if m == OBJ_KEY_START:
            return (_WS | frozenset(b'"'), False)
        if m == OBJ_COLON:
            return (_WS | frozenset(b":"), False)
        if m == ARR_FIRST:
            return (_WS | _DIGITS | frozenset(b'{["-tfn]'), False)
        raise AssertionError(m)

    def _close_set(self) -> frozenset:
        """Bytes legal once the current value is complete: whitespace plus
        the container's delimiters (nothing but ws/EOS at top level)."""
        if not self.stack:
            return _WS
        return _WS | (frozenset(b",}") if self.stack[-1] == 0x7B
                      else frozenset(b",]"))

    def _eos_ok(self) -> bool:
        return not self.stack

    # ---------------------------------------------------------------- push

    def push(self, b: int) -> None:
        m = self.mode
        if m in (INT, INT_ZERO, FRAC, EXP) and (b in b",}]" or b in _WS):
            # number ended by a delimiter: close the number, re-dispatch
            self.mode = AFTER
            if b in _WS:
                return
            self._push_after(b)
            return
        if m == VALUE or m == ARR_FIRST:
            if b in _WS:
                return
            if m == ARR_FIRST and b == 0x5D:          # ']'
                self.stack.pop()
                self.mode = AFTER
                return
            self._start_value(b)
            return
        if m == STRING:
            if b == 0x22:                              # closing '"'
                if self._key:
                    self._key = False
                    self.mode = OBJ_COLON
                else:
                    self.mode = AFTER
            elif b == 0x5C:
                self.mode = STR_ESC
            return
        if m == STR_ESC:
            self.mode = STR_U1 if b == 0x75 else STRING
            return
        if m in (STR_U1, STR_U2, STR_U3):
            self.mode = m + 1
            return
        if m == STR_U4:
            self.mode = STRING
            return
        if m == LITERAL:
            self._lit = self._lit[1:]
            if not self._lit:
                self.mode = AFTER
            return
        if m == INT_NEG:
            self.mode = INT_ZERO if b == 0x30 else INT
            return
        if m in (INT, INT_ZERO):
            if b == 0x2E:
                self.mode = FRAC_START
            elif b in b"eE":
                self.mode = EXP_START
            return                                     # digit: stay
        if m == FRAC_START:
            self.mode = FRAC
            return
        if m == FRAC:
            if b in b"eE":
                self.mode = EXP_START
            return
        if m == EXP_START:
            self.mode = EXP_SIGNED if b in b"+-" else EXP
            return
        if m == EXP_SIGNED:
            self.mode = EXP
            return
        if m == EXP:
            return                                     # digit: stay
        if m == AFTER:
            if b in _WS:
                return
            self._push_after(b)
            return
        if m == OBJ_FIRST:
            if b in _WS:
                return
            if b == 0x7D:                              # '}'
                self.stack.pop()
                self.mode = AFTER
            else:                                      # '"' key
                self._key = True
                self.mode = STRING
            return
        if m == OBJ_KEY_START:
            if b in _WS:
                return
            self._key = True
            self.mode = STRING
            return
        if m == OBJ_COLON:
            if b in _WS:
                return
            self.mode = VALUE
            return
        raise AssertionError((m, b))

    def _start_value(self, b: int) -> None:
        if b == 0x7B:                                  # '{'
            self.stack.append(b)
            self.mode = OBJ_FIRST
        elif b == 0x5B:                                # '['
            self.stack.append(b)
            self.mode = ARR_FIRST
        elif b == 0x22:
            self._key = False
            self.mode = STRING
        elif b == 0x2D:
            self.mode = INT_NEG
        elif b == 0x30:
            self.mode = INT_ZERO
        elif b in _DIGITS:
            self.mode = INT
        elif b in _LITERALS:
            self._lit = _LITERALS[b]
            self.mode = LITERAL
        else:
            raise ValueError(f"illegal value start byte {b!r}")

    def _push_after(self, b: int) -> None:
        """Delimiter after a complete value: ',', '}', ']'."""
        if not self.stack:
            raise ValueError("trailing data after top-level value")
        top = self.stack[-1]
        if b == 0x2C:                                  # ','
            self.mode = OBJ_KEY_START if top == 0x7B else VALUE
        elif b == 0x7D and top == 0x7B:
            self.stack.pop()
            self.mode = AFTER
        elif b == 0x5D and top == 0x5B:
            self.stack.pop()
            self.mode = AFTER
        else:
            raise ValueError(f"illegal delimiter {b!r}")

    # -------------------------------------------------- token-level helpers

    def accepts(self, bs: bytes) -> bool:
        """Would pushing this byte string keep the document legal?
        State is saved/restored — the automaton does NOT advance. Used for
        token-level constrained decoding with multi-byte tokenizers."""
        saved = (list(self.stack), self.mode, self._lit, self._key)
        try:
            for b in bs:
                if b not in self.allowed()[0]:
                    return False
                self.push(b)
            return True
        finally:
            self.stack = list(saved[0])
            self.mode, self._lit, self._key = saved[1], saved[2], saved[3]

    def push_bytes(self, bs: bytes) -> None:
        for b in bs:
            self.push(b)


def mask_logits_token_level(row_logits, constraint, tokenizer,
                            eos_id: int) -> None:
    """format="json" enforcement for MULTI-BYTE tokenizers (HF BPE): a
    token is legal iff its raw bytes keep the automaton legal. Checking
    the whole vocab per step is wasteful, so candidates are checked
    lazily in descending-logit order (top-256 then top-4096 — a legal
    token outside the window could never win the sample anyway once the
    window's tokens are unmasked); if nothing in the window is legal the
    single-byte fallback tokens of the byte-level alphabet guarantee
    progress. In-place; one host sync per constrained row per step."""
    import torch
    allowed_bytes, eos_ok = constraint.allowed()
    v = row_logits.shape[0]
    legal: list[int] = []
    for k in (256, 4096):
        ids = row_logits.topk(min(k, v)).indices.tolist()
        for tid in ids:
            if tid == eos_id:
                continue
            bs = tokenizer.token_bytes(tid)
            if bs and constraint.accepts(bs):
                legal.append(tid)
        if legal or k >= v:
            break
    if not legal:
        for b in allowed_bytes:
            tid = tokenizer.byte_token_id(b)
            if tid is not None:
                legal.append(tid)
    if eos_ok:
        legal.append(eos_id)
    keep = torch.full_like(row_logits, float("-inf"))
    if legal:
        ix = torch.tensor(sorted(set(legal)), dtype=torch.long,
                          device=row_logits.device)
        keep[ix] = row_logits[ix]
    row_logits.copy_(keep)
