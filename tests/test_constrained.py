"""json_schema constrained decoding: FSM unit tests + engine conformance on
the synthetic tokenizer (reference carries json_schema on
SamplingParams, sampling/sampling_params.py:25)."""

import json

import pytest
import torch

from parallax_amd.server.constrained import GrammarMatcher, JsonSchemaFSM

SCHEMA = {
    "type": "object",
    "properties": {
        "name": {"type": "string"},
        "age": {"type": "integer"},
        "tags": {"type": "array", "items": {"type": "string"},
                 "minItems": 1, "maxItems": 3},
    },
}


def accepts(fsm, text):
    st = fsm.advance_str(fsm.initial(), text)
    return st is not None and fsm.is_complete(st)


def test_fsm_accepts_and_rejects():
    fsm = JsonSchemaFSM(SCHEMA)
    assert accepts(fsm, '{"name":"bob","age":42,"tags":["x"]}')
    assert accepts(fsm, '{"name":"a\\"b","age":-7,"tags":["a","b","c"]}')
    assert not accepts(fsm, '{"name":"bob","age":42,"tags":[]}')  # minItems
    assert not accepts(fsm, '{"name":"bob","age":4.5,"tags":["x"]}')  # int
    assert not accepts(fsm, '{"age":1,"name":"x","tags":["x"]}')  # prop order
    assert not accepts(fsm, '{"name":"bob","age":01,"tags":["x"]}')  # lead 0


def test_fsm_scalar_types():
    f = JsonSchemaFSM({"type": "object", "properties": {
        "k": {"enum": ["alpha", "beta"]},
        "v": {"type": "number"},
        "b": {"type": "boolean"},
        "z": {"type": "null"},
    }})
    assert accepts(f, '{"k":"beta","v":-1.5e3,"b":false,"z":null}')
    assert accepts(f, '{"k":"alpha","v":0.25,"b":true,"z":null}')
    assert not accepts(f, '{"k":"gamma","v":1,"b":true,"z":null}')
    assert not accepts(f, '{"k":"alpha","v":1,"b":maybe,"z":null}')


def test_masked_random_walk_conforms():
    """Sampling uniformly from the allowed-token mask always ends in a valid
    document for the schema."""
    import random

    vocab = [""] * 3 + [chr(c) for c in range(32, 127)]
    rng = random.Random(7)
    for trial in range(5):
        m = GrammarMatcher(json.dumps(SCHEMA), vocab, eos_ids=[2])
        out = []
        for _ in range(600):
            m.catch_up(out)
            ids = m.allowed_ids()
            assert ids
            if ids == [2]:
                break
            out.append(rng.choice(ids))
        else:
            pytest.fail("walk did not terminate")
        obj = json.loads("".join(vocab[t] for t in out))
        assert isinstance(obj["age"], int)
        assert 1 <= len(obj["tags"]) <= 3
        assert set(obj) == {"name", "age", "tags"}


def test_engine_schema_conformance_synthetic_tokenizer():
    """End-to-end: a random-weight model forced through the grammar mask emits
    schema-conformant JSON on the synthetic tokenizer."""
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams
    from parallax_amd.server.tokenizer_util import TokenizerWrapper

    tok = TokenizerWrapper(vocab_size=512)
    vocab = tok.vocab_strings()
    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=512, hidden_size=64,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
        intermediate_size=128, max_position_embeddings=512,
        eos_token_ids=[tok.eos_token_id],
    )
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=64,
                                 dtype=torch.float32), random_weights=True)
    eng.set_grammar_vocab(vocab)
    schema = json.dumps({"type": "object", "properties": {
        "a": {"type": "integer"}, "b": {"type": "boolean"}}})
    sp = SamplingParams(temperature=1.0, max_new_tokens=120,
                        json_schema=schema)
    out = eng.generate([[5, 9, 13]], [sp])
    toks = list(out.values())[0]
    text = "".join(
        vocab[t] for t in toks if t != tok.eos_token_id and t < len(vocab)
    )
    obj = json.loads(text)
    assert isinstance(obj.get("a"), int) and isinstance(obj.get("b"), bool)


def test_optional_properties():
    """A "required" subset lets the model skip optional keys (in order) but
    never a required one, and close only once all required keys are done."""
    import json

    from parallax_amd.server.constrained import JsonSchemaFSM

    fsm = JsonSchemaFSM(json.loads(json.dumps({
        "type": "object",
        "properties": {"a": {"type": "integer"}, "b": {"type": "boolean"},
                       "c": {"type": "string"}},
        "required": ["a", "c"],
    })))

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    assert accepts('{"a":1,"b":true,"c":"x"}')   # all keys
    assert accepts('{"a":1,"c":"x"}')            # optional b skipped
    assert not accepts('{"a":1}')                # required c missing
    assert not accepts('{"b":true,"c":"x"}')     # required a skipped
    assert not accepts('{"c":"x","a":1}')        # order violated
    assert not accepts('{"a":1,"b":true}')       # closed before required c
    assert not accepts('{"a":1,,"c":"x"}')


def test_all_optional_object():
    import json

    from parallax_amd.server.constrained import JsonSchemaFSM

    fsm = JsonSchemaFSM({"type": "object", "required": [],
                         "properties": {"x": {"type": "integer"}}})

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    assert accepts("{}")
    assert accepts('{"x":7}')
    assert not accepts('{"x":}')
    # round-trip with python's parser for everything the FSM accepts
    for t in ("{}", '{"x":7}'):
        json.loads(t)


def test_anyof_union():
    """anyOf/oneOf: the automaton accepts any branch (nullable fields are
    the common case in tool schemas)."""
    from parallax_amd.server.constrained import JsonSchemaFSM

    fsm = JsonSchemaFSM({
        "type": "object",
        "properties": {
            "v": {"anyOf": [{"type": "integer"}, {"type": "null"}]},
            "w": {"oneOf": [{"type": "string"},
                            {"type": "array", "items": {"type": "integer"}}]},
        },
    })

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    assert accepts('{"v":3,"w":"hi"}')
    assert accepts('{"v":null,"w":[1,2]}')
    assert accepts('{"v":-7,"w":[]}')
    assert not accepts('{"v":true,"w":"hi"}')   # neither branch
    assert not accepts('{"v":3,"w":7}')


def test_union_number_boundary():
    """A union branch completing at a delimiter hands the char onward."""
    from parallax_amd.server.constrained import JsonSchemaFSM

    fsm = JsonSchemaFSM({"type": "array",
                         "items": {"anyOf": [{"type": "number"},
                                             {"type": "boolean"}]}})

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    assert accepts("[1.5,true,2]")
    assert accepts("[]")
    assert not accepts("[1.5,]")
    assert not accepts("[null]")


def test_recursive_ref_schema():
    """$defs + $ref, including self-reference (a binary-tree schema)."""
    from parallax_amd.server.constrained import JsonSchemaFSM

    fsm = JsonSchemaFSM({
        "$defs": {
            "tree": {
                "type": "object",
                "properties": {
                    "v": {"type": "integer"},
                    "kids": {"type": "array",
                             "items": {"$ref": "#/$defs/tree"},
                             "maxItems": 2},
                },
                "required": ["v"],
            }
        },
        "$ref": "#/$defs/tree",
    })

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    assert accepts('{"v":1}')
    assert accepts('{"v":1,"kids":[{"v":2},{"v":3,"kids":[{"v":4}]}]}')
    assert not accepts('{"kids":[]}')          # required v missing
    assert not accepts('{"v":1,"kids":[5]}')   # item not a tree


def test_free_form_object_and_json_object_mode():
    """additionalProperties (arbitrary keys) + the json_object schema used by
    response_format={"type": "json_object"}."""
    import json

    from parallax_amd.server.constrained import JsonSchemaFSM
    from parallax_amd.server.sampling_params import ANY_JSON_OBJECT_SCHEMA

    fsm = JsonSchemaFSM(ANY_JSON_OBJECT_SCHEMA)

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    for t in ('{}', '{"a":1}', '{"x":{"y":[1,"two",null,true]},"z":-3.5}',
              '{"k":"v","k2":[{"deep":{}}]}'):
        assert accepts(t), t
        json.loads(t)  # everything accepted is valid JSON
    for t in ('[]', '"str"', '{"a":}', '{:1}', '{"a" 1}'):
        assert not accepts(t), t


def test_json_object_response_format_plumbing():
    from parallax_amd.server.sampling_params import SamplingParams

    sp = SamplingParams.from_openai(
        {"response_format": {"type": "json_object"}})
    assert sp.json_schema and "additionalProperties" in sp.json_schema


def test_string_length_bounds():
    from parallax_amd.server.constrained import JsonSchemaFSM

    fsm = JsonSchemaFSM({"type": "object", "properties": {
        "s": {"type": "string", "minLength": 2, "maxLength": 4}}})

    def accepts(text):
        st = fsm.advance_str(fsm.initial(), text)
        return st is not None and fsm.is_complete(st)

    assert accepts('{"s":"ab"}')
    assert accepts('{"s":"abcd"}')
    assert accepts('{"s":"a\\n"}')     # escape counts as one char
    assert not accepts('{"s":"a"}')    # too short
    assert not accepts('{"s":"abcde"}')  # too long


@pytest.mark.parametrize("schema,check", [
    ({"type": "object",
      "properties": {"a": {"type": "integer"}, "b": {"type": "boolean"},
                     "c": {"type": "string", "maxLength": 6}},
      "required": ["a"]},
     lambda o: isinstance(o.get("a"), int)
     and set(o) <= {"a", "b", "c"}),
    ({"type": "object",
      "properties": {"v": {"anyOf": [{"type": "null"},
                                     {"type": "number"}]}}},
     lambda o: o["v"] is None or isinstance(o["v"], (int, float))),
    ({"$defs": {"t": {"type": "object",
                      "properties": {"n": {"type": "integer"},
                                     "k": {"type": "array",
                                           "items": {"$ref": "#/$defs/t"},
                                           "maxItems": 2}},
                      "required": ["n"]}},
      "$ref": "#/$defs/t"},
     lambda o: isinstance(o["n"], int)),
    ({"type": "object",
      "additionalProperties": {"type": "integer"}},
     lambda o: all(isinstance(v, int) for v in o.values())),
])
def test_masked_walk_conforms_across_features(schema, check):
    """Random masked walks terminate in valid JSON for every schema feature
    family (optional keys, unions, recursion, free-form objects)."""
    import random

    vocab = [""] * 3 + [chr(c) for c in range(32, 127)]
    rng = random.Random(11)
    for trial in range(4):
        m = GrammarMatcher(json.dumps(schema), vocab, eos_ids=[2])
        out = []
        for _ in range(800):
            m.catch_up(out)
            ids = m.allowed_ids()
            assert ids, "dead end in the automaton"
            if ids == [2]:
                break
            out.append(rng.choice(ids))
        else:
            pytest.fail("walk did not terminate")
        obj = json.loads("".join(vocab[t] for t in out))
        assert check(obj)
