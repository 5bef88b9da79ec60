"""Property-based checks of the oracle's GF arithmetic (hypothesis):
field laws that any correct GF(2^8)/0x11d and GF(2^16)/0x1100B
implementation must satisfy, independent of tables or construction."""
from hypothesis import given, settings
from hypothesis import strategies as st

import oracle

u8 = st.integers(min_value=0, max_value=255)
u16 = st.integers(min_value=0, max_value=65535)


@settings(max_examples=300, deadline=None)
@given(u8, u8, u8)
def test_gf8_field_laws(a, b, c):
    m = oracle.gf_mul
    assert m(a, b) == m(b, a)
    assert m(a, m(b, c)) == m(m(a, b), c)
    assert m(a, b ^ c) == m(a, b) ^ m(a, c)
    assert m(a, 1) == a and m(a, 0) == 0


@settings(max_examples=300, deadline=None)
@given(u8)
def test_gf8_inverse_and_frobenius(a):
    if a:
        assert oracle.gf_mul(a, oracle.gf_inv(a)) == 1
    # Frobenius: squaring is additive in characteristic 2
    for b in (1, 2, 0x1d):
        assert oracle.gf_mul(a ^ b, a ^ b) == \
            oracle.gf_mul(a, a) ^ oracle.gf_mul(b, b)


@settings(max_examples=200, deadline=None)
@given(u16, u16, u16)
def test_gf16_field_laws(a, b, c):
    m = oracle.gf16_mul
    assert m(a, b) == m(b, a)
    assert m(a, m(b, c)) == m(m(a, b), c)
    assert m(a, b ^ c) == m(a, b) ^ m(a, c)
    assert m(a, 1) == a and m(a, 0) == 0


def test_gf8_embeds_in_operations():
    """The generator 2 has order 255 in GF(2^8)/0x11d and 65535 in
    GF(2^16)/0x1100B (primitive polynomials)."""
    x, seen = 1, set()
    for _ in range(255):
        seen.add(x)
        x = oracle.gf_mul(x, 2)
    assert x == 1 and len(seen) == 255
    x, n = 1, 0
    while True:
        x = oracle.gf16_mul(x, 2)
        n += 1
        if x == 1:
            break
    assert n == 65535


@settings(max_examples=60, deadline=None)
@given(st.sampled_from(["reed_sol_van", "cauchy", "jerasure_reed_sol_van"]),
       st.integers(min_value=2, max_value=12),
       st.integers(min_value=1, max_value=4),
       st.integers(min_value=1, max_value=8),
       st.randoms(use_true_random=False))
def test_oracle_roundtrip_random_shapes(tech, k, m, blocks, rnd):
    """Encode -> erase up to m chunks -> decode reproduces the stripe,
    over randomized shapes and erasure patterns (hypothesis-driven CPU
    companion of the GPU fuzz sweep)."""
    import numpy as np
    C = 16 * blocks
    rng = np.random.default_rng(rnd.randrange(2**32))
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    par = oracle.encode(tech, k, m, data)
    n = k + m
    ne = rnd.randrange(1, m + 1)
    erased = sorted(rnd.sample(range(n), ne))
    chunks = [d.copy() for d in data] + [p.copy() for p in par]
    ref = data + par
    present = np.ones(n, np.uint8)
    for e in erased:
        present[e] = 0
        chunks[e][:] = 0
    oracle.decode(tech, k, m, chunks, present)
    for i in range(n):
        assert (chunks[i] == ref[i]).all(), (tech, k, m, erased, i)
