"""Property-based object-layer check: random write/read/truncate/flush
sequences must match a plain bytearray model byte-for-byte (SURVEY §5 unit
tier, hypothesis)."""

import os
import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from edgehip.lib import Engine

MB = 1 << 20
SPAN = 6 * MB  # keep offsets small enough to run fast, big enough for chunks


def rand_payload(rng, n):
    return bytes(rng.randrange(256) for _ in range(0)) or (b"%d" % rng.getrandbits(32)) * (n // 10 + 1)


ops = st.lists(
    st.one_of(
        st.tuples(st.just("write"), st.integers(0, SPAN), st.integers(1, MB)),
        st.tuples(st.just("read"), st.integers(0, SPAN + MB), st.integers(1, MB)),
        st.tuples(st.just("truncate"), st.integers(0, SPAN), st.just(0)),
        st.tuples(st.just("flush"), st.just(0), st.just(0)),
        st.tuples(st.just("punch"), st.integers(0, SPAN), st.integers(1, 2 * MB)),
        st.tuples(st.just("selfcopy"), st.integers(0, SPAN), st.integers(1, 2 * MB)),
    ),
    min_size=1,
    max_size=12,
)


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(ops=ops, seed=st.integers(0, 2**31))
def test_object_matches_model(ops, seed):
    import random
    rng = random.Random(seed)
    with Engine("engine=host,fp=blake3,compress=lz4,dedup=1") as eng, \
            eng.object() as o:
        model = bytearray()
        for op, a, b in ops:
            if op == "write":
                data = bytes(rng.getrandbits(8) for _ in range(min(b, 4096)))
                data = (data * (b // len(data) + 1))[:b] if data else b"\x00" * b
                o.write(a, data)
                if len(model) < a + b:
                    model.extend(b"\x00" * (a + b - len(model)))
                model[a:a + b] = data
            elif op == "read":
                got = o.read(a, b)
                want = bytes(model[a:a + b])
                assert got == want, f"read({a},{b}) mismatch"
            elif op == "truncate":
                o.truncate(a)
                if a <= len(model):
                    del model[a:]
                else:
                    model.extend(b"\x00" * (a - len(model)))
            elif op == "flush":
                o.flush()
            elif op == "punch":
                o.punch_hole(a, b)
                end = min(a + b, len(model))
                if a < len(model):
                    model[a:end] = b"\x00" * (end - a)
            elif op == "selfcopy":
                dst = (a + 2 * MB) % (SPAN + MB)
                got = o.copy_range(o, a, dst, b)
                want = min(b, max(0, len(model) - a)) if a < len(model) else 0
                assert got == want, f"selfcopy({a},{dst},{b}) -> {got} != {want}"
                if got:
                    if len(model) < dst + got:
                        model.extend(b"\x00" * (dst + got - len(model)))
                    model[dst:dst + got] = model[a:a + got]
        assert o.size == len(model)
        assert o.read(0, len(model) + 10) == bytes(model)


@pytest.mark.skipif(not os.environ.get("EDGEHIP_DEEP"),
                    reason="deep fuzz: set EDGEHIP_DEEP=1 (minutes, not CI-default)")
@settings(max_examples=400, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(ops=st.lists(
    st.one_of(
        st.tuples(st.just("write"), st.integers(0, SPAN), st.integers(1, MB)),
        st.tuples(st.just("read"), st.integers(0, SPAN + MB), st.integers(1, MB)),
        st.tuples(st.just("truncate"), st.integers(0, SPAN), st.just(0)),
        st.tuples(st.just("flush"), st.just(0), st.just(0)),
        st.tuples(st.just("punch"), st.integers(0, SPAN), st.integers(1, 2 * MB)),
        st.tuples(st.just("selfcopy"), st.integers(0, SPAN), st.integers(1, 2 * MB)),
    ), min_size=4, max_size=30), seed=st.integers(0, 2**31))
def test_object_matches_model_deep(ops, seed):
    """Long-sequence variant (found the punch-at-0 bug that 25 short examples
    missed). Run with EDGEHIP_DEEP=1."""
    test_object_matches_model.hypothesis.inner_test(ops, seed)
