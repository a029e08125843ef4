"""json_schema constrained decoding: a character-level pushdown automaton
compiled from a JSON-schema subset, driving token-level logit masks in the
sampler.

Reference parity: the reference carries `json_schema` on SamplingParams
(src/parallax/server/sampling/sampling_params.py:25) and delegates enforcement
to its backends; here enforcement is native. Schema subset: object (properties generated in
schema order; a "required" list makes the others optional — the automaton
lets the model skip any run of optional keys but never a required one),
string, integer, number, boolean, null, enum (distinct literals), array
(items + minItems/maxItems), anyOf/oneOf unions (nullable fields),
$ref/$defs including recursive references, and arbitrary nesting of those.

Design: states are immutable tuples (a stack of frames, top last), so the
per-state token mask is cacheable. `advance` consumes one character;
delimiters that terminate numbers are re-processed against the remaining
stack, which is what makes `{"a":12}`-style adjacency work. Token masking
tests each vocab string against the automaton; masks are cached per
(schema, state) so steady-state decoding costs one dict lookup.
"""

from __future__ import annotations

import json
from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

State = Tuple  # tuple of frames; () == complete

_DIGITS = set("0123456789")


class JsonSchemaFSM:
    def __init__(self, schema: dict):
        # flatten schema nodes into a list so frames can reference them by id
        self.nodes: List[dict] = []
        defs = schema.get("$defs") or schema.get("definitions") or {}
        # reserve ids for every named def first so $ref cycles (recursive
        # schemas: trees, linked structures) resolve to stable ids
        self._defs: Dict[str, int] = {}
        for name in defs:
            self._defs[name] = len(self.nodes)
            self.nodes.append({})
        for name, sub in defs.items():
            self._compile(sub, nid=self._defs[name])
        self.root = self._compile(schema)

    def _compile(self, schema: dict, nid: Optional[int] = None) -> int:
        if "$ref" in schema:
            ref = schema["$ref"]
            name = ref.rsplit("/", 1)[-1]
            if name not in self._defs:
                raise ValueError(f"unresolvable $ref: {ref!r}")
            return self._defs[name]
        if nid is None:
            nid = len(self.nodes)
            self.nodes.append({})
        node: Dict = {}
        if "anyOf" in schema or "oneOf" in schema:
            branches = schema.get("anyOf") or schema.get("oneOf")
            node = {"kind": "union",
                    "children": tuple(self._compile(b) for b in branches)}
        elif "enum" in schema:
            node = {"kind": "enum",
                    "options": tuple(json.dumps(v, separators=(",", ":"))
                                     for v in schema["enum"])}
        else:
            t = schema.get("type", "object")
            if t == "object":
                props = schema.get("properties", {})
                order = [k for k in props]
                if not order and isinstance(
                        schema.get("additionalProperties"), dict):
                    # free-form object: arbitrary string keys, value schema
                    # from additionalProperties (this is what makes
                    # response_format json_object expressible)
                    node = {"kind": "freeobject",
                            "value": self._compile(
                                schema["additionalProperties"])}
                    self.nodes[nid] = node
                    return nid
                # no "required" list = every property required (strict-mode
                # default; also the pre-existing behavior)
                req = frozenset(schema["required"]) if "required" in schema \
                    else frozenset(order)
                node = {
                    "kind": "object",
                    "keys": tuple(order),
                    "children": tuple(self._compile(props[k]) for k in order),
                    "required": req,
                }
            elif t == "array":
                node = {
                    "kind": "array",
                    "item": self._compile(schema.get("items", {"type": "string"})),
                    "min": int(schema.get("minItems", 0)),
                    "max": int(schema.get("maxItems", 1 << 30)),
                }
            elif t == "string":
                node = {"kind": "string",
                        "minlen": int(schema.get("minLength", 0)),
                        "maxlen": int(schema.get("maxLength", 1 << 30))}
            elif t in ("integer", "number", "boolean", "null"):
                node = {"kind": t}
            else:
                raise ValueError(f"unsupported schema type: {t!r}")
        self.nodes[nid] = node
        return nid

    # -- state construction ------------------------------------------------------

    def initial(self) -> State:
        return (("val", self.root),)

    def is_complete(self, state: State) -> bool:
        state = self._expand(state)
        if len(state) == 0:
            return True
        top, rest = state[-1], state[:-1]
        if top[0] == "union":
            return any(self.is_complete(sub) for sub in top[1]) \
                and self.is_complete(rest)
        if top[0] in ("int", "num") and top[1] in ("z", "int", "frac", "exp"):
            # a number is complete at any digit boundary
            return self.is_complete(rest)
        return False

    # -- the automaton ----------------------------------------------------------

    def _expand(self, state: State) -> State:
        """Expand a top-of-stack val frame into concrete frames (no input)."""
        while state and state[-1][0] == "val":
            nid = state[-1][1]
            node = self.nodes[nid]
            rest = state[:-1]
            kind = node["kind"]
            if kind == "enum":
                state = rest + (("alt", node["options"], ""),)
            elif kind == "string":
                # bounded strings carry a length counter; unbounded ones use
                # the counterless frame so every string state is shared (one
                # cached mask covers all inside-string positions)
                if node["minlen"] or node["maxlen"] < (1 << 30):
                    state = rest + (("strn", node["minlen"], node["maxlen"],
                                     0), ("lit", '"', 0))
                else:
                    state = rest + (("str",), ("lit", '"', 0))
            elif kind == "integer":
                state = rest + (("int", "start"),)
            elif kind == "number":
                state = rest + (("num", "start"),)
            elif kind == "boolean":
                state = rest + (("alt", ("true", "false"), ""),)
            elif kind == "null":
                state = rest + (("lit", "null", 0),)
            elif kind == "object":
                state = rest + (("obj", nid, 0, "open"),)
            elif kind == "freeobject":
                state = rest + (("fobj", nid, "open"),)
            elif kind == "union":
                subs = tuple((("val", c),) for c in node["children"])
                state = rest + (("union", subs),)
            elif kind == "array":
                state = rest + (("arr", nid, 0, "item_or_close"),
                                ("lit", "[", 0))
            else:  # pragma: no cover
                raise AssertionError(kind)
        return state

    def advance(self, state: State, ch: str) -> Optional[State]:
        state = self._expand(state)
        if not state:
            return None  # complete: no more characters accepted
        top = state[-1]
        rest = state[:-1]
        tag = top[0]

        if tag == "lit":
            _, text, i = top
            if ch != text[i]:
                return None
            if i + 1 == len(text):
                return rest
            return rest + (("lit", text, i + 1),)

        if tag == "str":
            if ch == '"':
                return rest
            if ch == "\\":
                return rest + (top, ("esc",))
            if ord(ch) < 0x20:
                return None
            return state

        if tag == "strn":
            _, mn, mx, n = top
            if ch == '"':
                return rest if n >= mn else None
            if ch == "\\":
                return rest + (("strn", mn, mx, n + 1), ("esc",)) \
                    if n < mx else None
            if ord(ch) < 0x20 or n >= mx:
                return None
            return rest + (("strn", mn, mx, n + 1),)

        if tag == "esc":
            if ch in '"\\/bfnrt':
                return rest
            return None

        if tag == "alt":
            _, options, prefix = top
            p = prefix + ch
            live = [o for o in options if o.startswith(p)]
            if not live:
                return None
            if p in live and len(live) == 1:
                return rest
            return rest + (("alt", options, p),)

        if tag in ("int", "num"):
            _, phase = top
            is_num = tag == "num"
            if phase == "start":
                if ch == "-":
                    return rest + ((tag, "int0"),)
                if ch == "0":
                    return rest + ((tag, "z"),)
                if ch in _DIGITS:
                    return rest + ((tag, "int"),)
                return None
            if phase == "int0":
                if ch == "0":
                    return rest + ((tag, "z"),)
                if ch in _DIGITS:
                    return rest + ((tag, "int"),)
                return None
            if phase in ("z", "int", "frac", "exp"):
                if ch in _DIGITS and phase != "z":
                    return rest + ((tag, phase),)
                if is_num and ch == "." and phase in ("z", "int"):
                    return rest + ((tag, "frac0"),)
                if is_num and ch in "eE" and phase in ("z", "int", "frac"):
                    return rest + ((tag, "exp0"),)
                if ch in _DIGITS and phase == "z":
                    return None  # no leading zeros
                # number complete: delimiter re-processed by the rest
                return self.advance(rest, ch)
            if phase == "frac0":
                return rest + ((tag, "frac"),) if ch in _DIGITS else None
            if phase == "exp0":
                if ch in "+-":
                    return rest + ((tag, "exp1"),)
                return rest + ((tag, "exp"),) if ch in _DIGITS else None
            if phase == "exp1":
                return rest + ((tag, "exp"),) if ch in _DIGITS else None
            return None

        if tag == "union":
            # nondeterministic branch embedded in one frame: advance every
            # live branch; a branch completing hands the character to the
            # enclosing context only when no branch can consume it
            # (maximal munch, same as the number frames)
            subs = top[1]
            new_subs = []
            for sub in subs:
                adv = self.advance(sub, ch)
                if adv is not None:
                    new_subs.append(adv)
            if new_subs:
                return rest + (("union", tuple(new_subs)),)
            if any(self.is_complete(sub) for sub in subs):
                return self.advance(rest, ch)
            return None

        if tag == "obj":
            _, nid, i, mode = top
            node = self.nodes[nid]
            keys, req = node["keys"], node["required"]
            may_close = not any(k in req for k in keys[i:])
            if mode == "open":
                if ch != "{":
                    return None
                return rest + (("obj", nid, 0, "key_or_close"),)
            if mode == "key_or_close":
                if ch == "}" and may_close:
                    return rest
                if i >= len(keys):
                    return None
                return self.advance(rest + (("objkey", nid, i, ""),), ch)
            if mode == "key":  # after a comma: a key MUST follow
                if i >= len(keys):
                    return None
                return self.advance(rest + (("objkey", nid, i, ""),), ch)
            if mode == "sep_or_close":
                if ch == "}" and may_close:
                    return rest
                if ch == "," and i < len(keys):
                    return rest + (("obj", nid, i, "key"),)
                return None
            return None

        if tag == "fobj":
            _, nid, mode = top
            node = self.nodes[nid]
            if mode == "open":
                return rest + (("fobj", nid, "key_or_close"),) \
                    if ch == "{" else None
            key_frames = (("fobj", nid, "sep_or_close"),
                          ("val", node["value"]), ("lit", ":", 0), ("str",))
            if mode == "key_or_close":
                if ch == "}":
                    return rest
                return rest + key_frames if ch == '"' else None
            if mode == "key":  # after a comma a key MUST follow
                return rest + key_frames if ch == '"' else None
            if mode == "sep_or_close":
                if ch == "}":
                    return rest
                if ch == ",":
                    return rest + (("fobj", nid, "key"),)
                return None
            return None

        if tag == "objkey":
            # lazily choose WHICH key comes next by matching characters
            # against the candidate quoted keys (schema order; a run of
            # optional keys may be skipped, a required key may not)
            _, nid, i, prefix = top
            node = self.nodes[nid]
            keys, req = node["keys"], node["required"]
            opts = []
            for j in range(i, len(keys)):
                opts.append((f'"{keys[j]}":', j))
                if keys[j] in req:
                    break
            p = prefix + ch
            live = [(o, j) for o, j in opts if o.startswith(p)]
            if not live:
                return None
            for o, j in live:
                # quoted forms are never prefixes of each other, so an exact
                # match is unique and final
                if o == p:
                    return rest + (("obj", nid, j + 1, "sep_or_close"),
                                   ("val", node["children"][j]))
            return rest + (("objkey", nid, i, p),)

        if tag == "arr":
            _, nid, n, expect = top
            node = self.nodes[nid]
            if expect == "item_or_close":
                if ch == "]" and n >= node["min"]:
                    return rest
                if n >= node["max"]:
                    return None
                nxt = rest + (("arr", nid, n + 1, "sep_or_close"),
                              ("val", node["item"]))
                return self.advance(nxt, ch)
            if expect == "sep_or_close":
                if ch == "]" and n >= node["min"]:
                    return rest
                if ch == "," and n < node["max"]:
                    return rest + (("arr", nid, n + 1, "sep_or_close"),
                                   ("val", node["item"]))
                return None
            return None

        return None  # pragma: no cover

    def advance_str(self, state: State, s: str) -> Optional[State]:
        for ch in s:
            state = self.advance(state, ch)
            if state is None:
                return None
        return self._expand(state)


class GrammarMatcher:
    """Per-request masking state: tracks consumed output tokens and yields the
    set of allowed next-token ids (EOS once the schema is satisfied). Masks
    are cached per (schema, state) across all requests."""

    _fsm_cache: Dict[str, JsonSchemaFSM] = {}
    # mask cache is LRU-bounded: recursive schemas produce unboundedly many
    # distinct automaton states over a long-lived server
    _mask_cache: "OrderedDict[Tuple[str, State], List[int]]" = OrderedDict()
    _mask_cache_max = 20000
    _trie_cache: Dict[int, tuple] = {}  # id(vocab) -> (vocab, trie root)

    def __init__(self, schema_json: str, vocab: List[str], eos_ids: List[int]):
        self.schema_json = schema_json
        fsm = self._fsm_cache.get(schema_json)
        if fsm is None:
            fsm = JsonSchemaFSM(json.loads(schema_json))
            self._fsm_cache[schema_json] = fsm
        self.fsm = fsm
        self.vocab = vocab
        self.eos_ids = [e for e in eos_ids if e is not None]
        self.state: Optional[State] = fsm.initial()
        self._consumed = 0

    def catch_up(self, output_token_ids: List[int]) -> None:
        for tid in output_token_ids[self._consumed:]:
            if self.state is None:
                break
            if tid in self.eos_ids:
                continue
            s = self.vocab[tid] if 0 <= tid < len(self.vocab) else ""
            nxt = self.fsm.advance_str(self.state, s)
            if nxt is not None:
                self.state = nxt
            # else: token was sampled outside the grammar (mask raced an
            # abort or vocab mismatch) — freeze rather than crash
        self._consumed = len(output_token_ids)

    def _trie(self):
        """Prefix trie over the vocab: computing a mask walks the trie,
        advancing the automaton per EDGE — a rejected prefix prunes its whole
        subtree, so a mask costs O(grammar-reachable prefixes), not
        O(vocab x token length). At 128k vocab the linear scan was a
        multi-hundred-ms stall per previously-unseen automaton state; the
        richer schemas (recursion, unions) make unseen states common."""
        entry = self._trie_cache.get(id(self.vocab))
        if entry is not None and entry[0] is self.vocab:
            return entry[1]
        root: Dict = {"ids": [], "kids": {}}
        for i, tok in enumerate(self.vocab):
            if not tok:
                continue
            node = root
            for ch in tok:
                node = node["kids"].setdefault(ch, {"ids": [], "kids": {}})
            node["ids"].append(i)
        self._trie_cache[id(self.vocab)] = (self.vocab, root)
        return root

    def allowed_ids(self) -> List[int]:
        if self.state is None:
            return list(self.eos_ids)
        if self.fsm.is_complete(self.state):
            return list(self.eos_ids)
        key = (self.schema_json, self.state)
        ids = self._mask_cache.get(key)
        if ids is None:
            ids = []
            adv = self.fsm.advance
            stack = [(self.state, self._trie())]
            while stack:
                st, node = stack.pop()
                ids.extend(node["ids"])
                for ch, child in node["kids"].items():
                    nxt = adv(st, ch)
                    if nxt is not None:
                        stack.append((nxt, child))
            ids.sort()
            self._mask_cache[key] = ids
            if len(self._mask_cache) > self._mask_cache_max:
                self._mask_cache.popitem(last=False)
        else:
            self._mask_cache.move_to_end(key)
        return ids
