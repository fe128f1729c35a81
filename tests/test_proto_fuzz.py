"""Wire-codec fuzz: random field values for every message type must
survive encode -> decode -> encode byte-identically (the .caffemodel
compat story rests on this codec)."""

import random

from poseidon_amd.proto import Message, spec


def _fill(msg: Message, rng: random.Random, depth: int) -> None:
    schema = spec.MESSAGES[msg.type_name]
    for fname, (num, ftype, label, default) in schema.items():
        if rng.random() < 0.4:
            continue
        rep = label == "rep"
        count = rng.randint(1, 3) if rep else 1
        for _ in range(count):
            if ftype in spec.MESSAGES:
                if depth <= 0:
                    continue
                sub = msg.add(fname) if rep else msg.ensure(fname)
                _fill(sub, rng, depth - 1)
                continue
            if ftype in spec.ENUMS:
                v = rng.choice(list(spec.ENUMS[ftype].values()))
            elif ftype == "bool":
                v = rng.random() < 0.5
            elif ftype in ("int32", "int64", "sint32"):
                v = rng.randint(-(1 << 20), 1 << 20)
            elif ftype in ("uint32", "uint64"):
                v = rng.randint(0, 1 << 21)
            elif ftype in ("float", "double"):
                v = rng.choice([0.0, 1.5, -2.25, 1e-7, 3e8])
            elif ftype in ("string", "bytes"):
                v = "s" + str(rng.randint(0, 999))
            else:
                continue
            if rep:
                getattr(msg, fname).append(v)
            else:
                setattr(msg, fname, v)


def test_roundtrip_every_message_type():
    rng = random.Random(1234)
    for tname in sorted(spec.MESSAGES):
        for trial in range(3):
            m = Message(tname)
            _fill(m, rng, depth=2)
            wire = m.encode()
            back = Message.decode(tname, wire)
            assert back.encode() == wire, f"{tname} trial {trial}"


def test_fuzz_cross_google():
    """Fuzzed messages encoded by OUR codec must parse with google.protobuf
    and re-serialize (deterministic) to the identical bytes."""
    import pytest
    pytest.importorskip("google.protobuf")
    from tests.test_proto import _build_google_pool
    classes = _build_google_pool()
    rng = random.Random(99)
    checked = 0
    for tname in sorted(spec.MESSAGES):
        if tname not in classes:
            continue
        for _ in range(3):
            m = Message(tname)
            _fill(m, rng, depth=2)
            wire = m.encode()
            g = classes[tname]()
            g.ParseFromString(wire)
            assert g.SerializeToString(deterministic=True) == wire, tname
            checked += 1
    assert checked >= 30
