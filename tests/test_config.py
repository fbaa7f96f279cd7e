import pytest

from dsin_amd.config import ConfigError, parse_string


def test_basic_values():
    cfg = parse_string("""
a = 1
b = 2*0.02  # arithmetic
c = (320, 960)
d = 'hello'
e = None
f = True
""")
    assert cfg.a == 1
    assert abs(cfg.b - 0.04) < 1e-12
    assert cfg.c == (320, 960)
    assert cfg.d == "hello"
    assert cfg.e is None
    assert cfg.f is True


def test_bare_identifier_is_string():
    cfg = parse_string("x = mae\ny = FIXED\n")
    assert cfg.x == "mae"
    assert cfg.y == "FIXED"


def test_constrain_ok_and_violation():
    cfg = parse_string("constrain m :: mse, mae\nm = mae\n")
    assert cfg.m == "mae"
    with pytest.raises(ConfigError):
        parse_string("constrain m :: mse, mae\nm = psnr\n")


def test_setattr_revalidates():
    cfg = parse_string("constrain m :: mse, mae\nm = mae\n")
    with pytest.raises(ConfigError):
        cfg.m = "nope"


def test_str_dump_roundtrips_keys():
    cfg = parse_string("a = 1\nb = 'x'\n")
    s = str(cfg)
    assert "a = 1" in s and "b = 'x'" in s


def test_clone_overrides():
    cfg = parse_string("a = 1\n")
    c2 = cfg.clone(a=5)
    assert c2.a == 5 and cfg.a == 1


def test_rejects_calls():
    with pytest.raises(ConfigError):
        parse_string("a = __import__('os')\n")


try:
    from hypothesis import given, settings, strategies as st
    HAVE_HYP = True
except ImportError:  # pragma: no cover
    HAVE_HYP = False


if HAVE_HYP:
    _keys = st.from_regex(r"[a-z][a-z0-9_]{0,15}", fullmatch=True)
    _scalars = st.one_of(
        st.integers(-10**6, 10**6),
        st.floats(allow_nan=False, allow_infinity=False, width=32),
        st.booleans(), st.none(),
        st.text(st.characters(whitelist_categories=("Ll", "Nd")),
                min_size=1, max_size=12),
    )
    _values = st.one_of(_scalars,
                        st.tuples(_scalars, _scalars),
                        st.lists(_scalars, min_size=1, max_size=4))

    @settings(max_examples=40, deadline=None)
    @given(st.dictionaries(_keys, _values, min_size=1, max_size=12))
    def test_dump_reparse_value_roundtrip(d):
        """The sidecar dump (configs_<name>.txt contract) must reparse to the
        exact same values for every supported value type."""
        cfg = parse_string("\n".join(f"{k} = {v!r}" for k, v in d.items()))
        cfg2 = parse_string(str(cfg))
        assert cfg2.as_dict() == cfg.as_dict() == d
