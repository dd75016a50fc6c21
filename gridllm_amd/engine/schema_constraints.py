"""JSON-Schema-constrained decoding: a schema-directed byte automaton.

Ollama enforces `format: {<json schema>}` through llama.cpp grammars; the
reference forwarded the object untouched (server/src/routes/ollama.ts:45,
openai.ts:637-642). Here the engine enforces a practical schema subset
directly, byte-by-byte, stacked on the plain JSON automaton
(constraints.py): masking keeps every sampled byte inside the schema, so
a completed generation is guaranteed schema-valid by construction.

Supported subset:
  type: object / array / string / number / integer / boolean / null
  object: properties, required, additionalProperties (False restricts
          keys to `properties` and lets masking FORCE the required keys;
          a dict supplies the schema for unknown keys)
  array:  items, minItems, maxItems
  string: minLength / maxLength (code-point counted)
  enum / const: any JSON values, matched by their compact serialization
  anyOf / oneOf: parallel branch automata (a byte is legal if any live
          branch accepts it)
  $ref:   local ('#/'-rooted) pointers, incl. RECURSIVE schemas (frames
          build lazily as the document descends)
Anything else (patterns, numeric bounds, type lists…) degrades that
subtree to the plain syntactic-JSON constraint.

Output is COMPACT JSON: insignificant whitespace is not offered to the
sampler (standard for grammar-constrained decoding; any JSON consumer
accepts it).

The interface matches JsonByteConstraint — allowed() -> (frozenset bytes,
eos_ok), push(byte), accepts(bytes), push_bytes — so both masking paths
(byte-exact and HF token-level, engine.py::_apply_constraint_mask) work
unchanged.
"""

from __future__ import annotations

import json
from typing import Optional

from gridllm_amd.engine import constraints as C

_STR_MODES = (C.STRING, C.STR_ESC, C.STR_U1, C.STR_U2, C.STR_U3, C.STR_U4)
_VALUE_START = frozenset(b'{["-0123456789tfn')
_NUM_START = frozenset(b"-0123456789")


def _deref(schema, root):
    """Resolve local $ref chains ('#/$defs/x', '#/definitions/x', any
    '#/'-rooted pointer). Lazy frame construction makes RECURSIVE schemas
    safe: a frame for a $ref subtree is only built when the document
    actually descends into it."""
    hops = 0
    while isinstance(schema, dict) and isinstance(schema.get("$ref"), str) \
            and schema["$ref"].startswith("#") and root is not None:
        hops += 1
        if hops > 32:
            return True                   # cyclic ref chain: degrade
        node = root
        for part in schema["$ref"].lstrip("#/").split("/"):
            if not part:
                continue
            part = part.replace("~1", "/").replace("~0", "~")
            if isinstance(node, dict) and part in node:
                node = node[part]
            else:
                return True               # unresolvable: degrade
        schema = node
    return schema


def _dumps(v) -> bytes:
    return json.dumps(v, separators=(",", ":"), ensure_ascii=False) \
        .encode("utf-8")


class _Gen:
    """Subtree constrained only syntactically (nested plain automaton),
    with optional first-byte restriction and banned bytes (used for
    string / number / integer / fallback)."""

    def __init__(self, first: Optional[frozenset] = None,
                 banned: frozenset = frozenset()):
        self.inner = C.JsonByteConstraint()
        self.started = False
        self.first = first
        self.banned = banned
        self.done = False                 # never self-closes; popped lazily

    def allowed(self) -> set:
        a = set(self.inner.allowed()[0])
        if self.inner.mode not in _STR_MODES:
            a -= C._WS                    # compact output
            a -= self.banned
        if not self.started and self.first is not None:
            a &= self.first
        return a

    def complete(self) -> bool:
        return self.started and self.inner.allowed()[1]

    def push(self, ctx, b: int):
        self.started = True
        self.inner.push(b)
        return None

    def child_done(self, f) -> None:  # pragma: no cover - leaf frame
        pass


class _Lit:
    """Byte-trie over candidate serializations (enum/const, true/false,
    null, restricted object keys)."""

    def __init__(self, options: list, values: Optional[list] = None):
        self.opts = list(options)
        self.vals = list(values) if values is not None else [None] * len(
            self.opts)
        self.pos = 0
        self.matched = None               # value of a fully-matched option
        self.done = False

    def allowed(self) -> set:
        return {o[self.pos] for o in self.opts if self.pos < len(o)}

    def complete(self) -> bool:
        return self.done or self.matched is not None

    def push(self, ctx, b: int):
        keep = [(o, v) for o, v in zip(self.opts, self.vals)
                if self.pos < len(o) and o[self.pos] == b]
        self.opts = [o for o, _ in keep]
        self.vals = [v for _, v in keep]
        self.pos += 1
        self.matched = None
        for o, v in keep:
            if len(o) == self.pos:
                self.matched = v if v is not None else o
        if all(len(o) == self.pos for o in self.opts):
            self.done = True
        return None

    def child_done(self, f) -> None:  # pragma: no cover - leaf frame
        pass


class _KeyAny:
    """An arbitrary object key, entered AFTER the opening quote; records
    the key text so the object can pick the right value schema."""

    def __init__(self):
        self.inner = C.JsonByteConstraint()
        self.inner.push(0x22)             # prime: inside a string
        self.raw = bytearray()            # content + closing quote
        self.done = False
        self.matched = None               # decoded key on completion

    def allowed(self) -> set:
        return set(self.inner.allowed()[0]) if not self.done else set()

    def complete(self) -> bool:
        return self.done

    def push(self, ctx, b: int):
        self.inner.push(b)
        self.raw.append(b)
        if self.inner.allowed()[1]:       # string closed -> complete value
            self.done = True
            try:
                self.matched = json.loads(
                    b'"' + bytes(self.raw), strict=False)
            except Exception:  # noqa: BLE001 — non-UTF8 content
                self.matched = bytes(self.raw[:-1]).decode(
                    "utf-8", errors="replace")
        return None

    def child_done(self, f) -> None:  # pragma: no cover - leaf frame
        pass


_ASCII_CONTENT = frozenset(b for b in range(0x20, 0x80)
                           if b not in (0x22, 0x5C))
# valid UTF-8 lead bytes (no overlongs via C0/C1, nothing past F4)
_LEADS = frozenset(range(0xC2, 0xF5))
_CONT_DEFAULT = (0x80, 0xBF)
# first-continuation range per lead (excludes overlongs, surrogates,
# and code points beyond U+10FFFF)
_FIRST_CONT = {0xE0: (0xA0, 0xBF), 0xED: (0x80, 0x9F),
               0xF0: (0x90, 0xBF), 0xF4: (0x80, 0x8F)}


class _Str:
    """String with minLength/maxLength: counts CODE POINTS. Content is
    restricted to VALID UTF-8 (lead-dependent continuation ranges — no
    overlongs/surrogates), so the automaton's count equals the decoded
    string's length; escapes count 1."""

    BODY, ESC, U1, U2, U3, U4 = range(6)

    _HEX = frozenset(b"0123456789abcdefABCDEF")
    _LOW2 = frozenset(b"cdefCDEF")       # 2nd hex digit of DC00-DFFF
    _HIGH2 = frozenset(b"89abAB")        # 2nd hex digit of D800-DBFF

    def __init__(self, schema: dict):
        self.min_l = int(schema.get("minLength") or 0)
        mx = schema.get("maxLength")
        self.max_l = int(mx) if mx is not None else None
        self.n = 0
        self.rem = 0                   # continuation bytes still owed
        self.crange = _CONT_DEFAULT    # range for the NEXT continuation
        self.state = None              # None = before opening quote
        self.done = False
        # surrogate-pair escapes: a \uD800-\uDBFF escape counts ONE code
        # point together with the \uDC00-\uDFFF escape that MUST follow
        # (round-1 advisor finding: pairs were counted as 2 and a lone
        # high surrogate could be force-closed into invalid text)
        self.pending_high = False      # inside a pair, low half owed
        self.hex0 = 0                  # first hex digit of current escape
        self.is_high = False           # current escape is a high surrogate

    def allowed(self) -> set:
        if self.done:
            return set()
        if self.state is None:
            return {0x22}
        if self.state == self.ESC:
            if self.pending_high:
                return {0x75}          # only \uDC00-\uDFFF may follow
            return frozenset(b'"\\/bfnrtu')
        if self.state == self.U1:
            if self.pending_high:
                return frozenset(b"dD")
            return self._HEX
        if self.state == self.U2:
            if self.hex0 in b"dD":
                if self.pending_high:
                    return self._LOW2          # low half of the pair
                # D800-DBFF starts a pair (allowed); lone DC00-DFFF banned
                return self._HEX - self._LOW2
            return self._HEX
        if self.state in (self.U3, self.U4):
            return self._HEX
        # BODY
        if self.pending_high:
            return {0x5C}              # the pair's low half is mandatory
        if self.rem:
            lo, hi = self.crange
            return set(range(lo, hi + 1))
        a: set = set()
        if self.max_l is None or self.n < self.max_l:
            a |= _ASCII_CONTENT | _LEADS
            a.add(0x5C)
        if self.n >= self.min_l:
            a.add(0x22)
        return a

    def complete(self) -> bool:
        return self.done

    def push(self, ctx, b: int):
        if self.state is None and b == 0x22:
            self.state = self.BODY
            return None
        if self.state == self.BODY:
            if self.rem:
                self.rem -= 1
                self.crange = _CONT_DEFAULT
                return None
            if b == 0x22:
                self.done = True
            elif b == 0x5C:
                self.state = self.ESC
                if not self.pending_high:   # pair counts once, at its start
                    self.n += 1
            else:
                self.n += 1
                if 0xC2 <= b <= 0xDF:
                    self.rem = 1
                elif 0xE0 <= b <= 0xEF:
                    self.rem = 2
                elif 0xF0 <= b <= 0xF4:
                    self.rem = 3
                self.crange = _FIRST_CONT.get(b, _CONT_DEFAULT)
            return None
        if self.state == self.ESC:
            self.state = self.U1 if b == 0x75 else self.BODY
            return None
        if self.state == self.U1:
            self.hex0 = b
            self.state = self.U2
            return None
        if self.state == self.U2:
            if self.hex0 in b"dD" and b in self._HIGH2:
                self.is_high = True     # this escape opens a pair
            else:
                self.is_high = False
            self.state = self.U3
            return None
        if self.state == self.U3:
            self.state = self.U4
            return None
        if self.state == self.U4:
            if self.pending_high:
                self.pending_high = False   # pair completed
            elif getattr(self, "is_high", False):
                self.pending_high = True    # low half now owed
            self.state = self.BODY
            return None
        raise ValueError(f"string: illegal byte {b!r}")

    def child_done(self, f) -> None:  # pragma: no cover - leaf frame
        pass


class _Obj:
    OPEN, KEY_OR_CLOSE, KEY, COLON, AFTER = range(5)

    def __init__(self, schema: dict, root=None):
        self.root = root
        self.props = dict(schema.get("properties") or {})
        self.required = set(schema.get("required") or [])
        for k in self.required:
            # required key without a property schema: give it a generic
            # one so the automaton can still offer (and force) it
            self.props.setdefault(k, True)
        addl = schema.get("additionalProperties", True)
        self.addl = addl
        self.seen: set = set()
        self.state = self.OPEN
        self.pending_key = None
        self.expect_key = False
        self.done = False

    def _open_keys(self) -> list:
        return [k for k in self.props if k not in self.seen]

    def _may_close(self) -> bool:
        return not (self.required - self.seen)

    def allowed(self) -> set:
        if self.done:
            return set()
        if self.state == self.OPEN:
            return {0x7B}
        if self.state == self.KEY_OR_CLOSE:
            a = set()
            if self._may_close():
                a.add(0x7D)
            if self._open_keys() or self.addl is not False:
                a.add(0x22)
            return a
        if self.state == self.KEY:
            return {0x22}                 # after ',' a key is mandatory
        if self.state == self.COLON:
            return {0x3A}
        if self.state == self.AFTER:
            a = set()
            if self._may_close():
                a.add(0x7D)
            if self._open_keys() or self.addl is not False:
                a.add(0x2C)
            return a
        return set()

    def complete(self) -> bool:
        return self.done

    def push(self, ctx, b: int):
        if self.state == self.OPEN and b == 0x7B:
            self.state = self.KEY_OR_CLOSE
            return None
        if self.state in (self.KEY_OR_CLOSE, self.AFTER) and b == 0x7D:
            self.done = True
            return None
        if self.state == self.AFTER and b == 0x2C:
            self.state = self.KEY
            return None
        if self.state in (self.KEY_OR_CLOSE, self.KEY) and b == 0x22:
            self.expect_key = True
            self.state = self.COLON
            if self.addl is False:
                opts = [_dumps(k)[1:] for k in self._open_keys()]
                return _Lit(opts, self._open_keys())
            return _KeyAny()
        if self.state == self.COLON and b == 0x3A:
            key = self.pending_key
            if key in self.props:
                sub = self.props[key]
            elif isinstance(self.addl, dict):
                sub = self.addl
            else:
                sub = True                # generic value
            self.state = self.AFTER
            return make_frame(sub, self.root)
        raise ValueError(f"object: illegal byte {b!r} in state {self.state}")

    def child_done(self, f) -> None:
        if self.expect_key:
            self.expect_key = False
            self.pending_key = f.matched
            self.seen.add(f.matched)


class _Arr:
    OPEN, FIRST, AFTER = range(3)

    def __init__(self, schema: dict, root=None):
        self.root = root
        self.items = schema.get("items", True)
        self.min_items = int(schema.get("minItems") or 0)
        mx = schema.get("maxItems")
        self.max_items = int(mx) if mx is not None else None
        self.count = 0
        self.state = self.OPEN
        self.done = False

    def allowed(self) -> set:
        if self.done:
            return set()
        if self.state == self.OPEN:
            return {0x5B}
        if self.state == self.FIRST:
            a = set(_first_bytes(self.items, self.root)) \
                if self.max_items is None or self.max_items > 0 else set()
            if self.min_items <= 0:
                a.add(0x5D)
            return a
        if self.state == self.AFTER:
            a = set()
            if self.count >= self.min_items:
                a.add(0x5D)
            if self.max_items is None or self.count < self.max_items:
                a.add(0x2C)
            return a
        return set()

    def complete(self) -> bool:
        return self.done

    def push(self, ctx, b: int):
        if self.state == self.OPEN and b == 0x5B:
            self.state = self.FIRST
            return None
        if self.state == self.FIRST and b == 0x5D and self.min_items <= 0:
            self.done = True
            return None
        if self.state == self.AFTER and b == 0x5D and \
                self.count >= self.min_items:
            self.done = True
            return None
        if self.state == self.FIRST:
            self.count += 1
            self.state = self.AFTER
            return (make_frame(self.items, self.root), b)  # replay b
        if self.state == self.AFTER and b == 0x2C:
            self.count += 1
            return make_frame(self.items, self.root)
        raise ValueError(f"array: illegal byte {b!r} in state {self.state}")

    def child_done(self, f) -> None:
        pass


class _Union:
    """anyOf / oneOf: parallel branch automata over the same byte stream.
    A byte is legal if ANY live branch accepts it; branches that refuse a
    consumed byte are dropped; the value completes when any surviving
    branch is complete."""

    def __init__(self, branches: list, root=None):
        self.branches = [JsonSchemaByteConstraint(s, root)
                         for s in branches]
        self.done = False

    def allowed(self) -> set:
        out: set = set()
        for br in self.branches:
            out |= br.allowed()[0]
        return out

    def complete(self) -> bool:
        return any(br.allowed()[1] for br in self.branches)

    def push(self, ctx, b: int):
        keep = []
        for br in self.branches:
            if b in br.allowed()[0]:
                br.push(b)
                keep.append(br)
        if not keep:
            raise ValueError(f"union: illegal byte {b!r}")
        self.branches = keep
        return None

    def child_done(self, f) -> None:  # pragma: no cover - leaf frame
        pass


def _first_bytes(schema, root=None) -> frozenset:
    """Bytes a value of this schema may start with (compact form)."""
    schema = _deref(schema, root)
    if not isinstance(schema, dict) or not schema:
        return _VALUE_START
    if "anyOf" in schema or "oneOf" in schema:
        out = frozenset()
        for s in schema.get("anyOf") or schema.get("oneOf") or []:
            out |= _first_bytes(s, root)
        return out or _VALUE_START
    if "enum" in schema or "const" in schema:
        vals = schema.get("enum", [schema.get("const")])
        return frozenset(_dumps(v)[0] for v in vals if _dumps(v))
    t = schema.get("type")
    if isinstance(t, list):
        out = frozenset()
        for ti in t:
            out |= _first_bytes({**schema, "type": ti}, root)
        return out
    if t == "object" or (t is None and "properties" in schema):
        return frozenset(b"{")
    if t == "array":
        return frozenset(b"[")
    if t == "string":
        return frozenset(b'"')
    if t in ("number", "integer"):
        return _NUM_START
    if t == "boolean":
        return frozenset(b"tf")
    if t == "null":
        return frozenset(b"n")
    return _VALUE_START


def make_frame(schema, root=None):
    """Schema node -> automaton frame. Unsupported constructs degrade to
    the plain syntactic constraint for that subtree."""
    schema = _deref(schema, root)
    if not isinstance(schema, dict) or not schema:
        return _Gen()
    if "anyOf" in schema or "oneOf" in schema:
        branches = [s for s in (schema.get("anyOf")
                                or schema.get("oneOf") or [])
                    if isinstance(s, (dict, bool))]
        if branches:
            return _Union(branches, root)
        return _Gen()
    if "enum" in schema or "const" in schema:
        vals = schema.get("enum", [schema.get("const")])
        return _Lit([_dumps(v) for v in vals], vals)
    t = schema.get("type")
    if isinstance(t, list):
        return _Gen(first=_first_bytes(schema))
    if t == "object" or (t is None and "properties" in schema):
        return _Obj(schema, root)
    if t == "array":
        return _Arr(schema, root)
    if t == "string":
        if schema.get("minLength") or schema.get("maxLength") is not None:
            return _Str(schema)
        return _Gen(first=frozenset(b'"'))
    if t == "integer":
        return _Gen(first=_NUM_START, banned=frozenset(b".eE"))
    if t == "number":
        return _Gen(first=_NUM_START)
    if t == "boolean":
        return _Lit([b"true", b"false"], [True, False])
    if t == "null":
        return _Lit([b"null"], [None])
    return _Gen()


class JsonSchemaByteConstraint:
    """Drop-in sibling of JsonByteConstraint, directed by a JSON schema."""

    def __init__(self, schema, root=None):
        if root is None and isinstance(schema, dict):
            root = schema
        self.stack = [make_frame(schema, root)]

    def allowed(self) -> tuple:
        out: set = set()
        for f in reversed(self.stack):
            out |= f.allowed()
            if not f.complete():
                break
        eos_ok = all(f.complete() for f in self.stack)
        return frozenset(out), eos_ok

    def push(self, b: int) -> None:
        while True:
            top = self.stack[-1]
            if b in top.allowed():
                res = top.push(self, b)
                if res is not None:
                    if isinstance(res, tuple):    # (child, replay byte)
                        child, rb = res
                        self.stack.append(child)
                        b = rb
                        continue
                    self.stack.append(res)
                break
            if top.complete() and len(self.stack) > 1:
                f = self.stack.pop()
                self.stack[-1].child_done(f)
                continue
            raise ValueError(f"schema: illegal byte {b!r}")
        while len(self.stack) > 1 and self.stack[-1].done:
            f = self.stack.pop()
            self.stack[-1].child_done(f)

    # same token-level helpers as JsonByteConstraint -------------------

    def accepts(self, bs: bytes) -> bool:
        saved = self._snapshot()
        try:
            for b in bs:
                if b not in self.allowed()[0]:
                    return False
                self.push(b)
            return True
        finally:
            self._restore(saved)

    def push_bytes(self, bs: bytes) -> None:
        for b in bs:
            self.push(b)

    # frames hold mutable state; deep-snapshot via pickle-free copy
    def _snapshot(self):
        import copy
        return copy.deepcopy(self.stack)

    def _restore(self, saved) -> None:
        self.stack = saved
