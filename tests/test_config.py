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
