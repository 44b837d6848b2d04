import pytest

from bigclam.config import BigClamConfig, k_grid


def test_validation_errors():
    with pytest.raises(ValueError, match="ls_steps"):
        BigClamConfig(ls_steps=16)
    with pytest.raises(ValueError, match="ls_steps"):
        BigClamConfig(ls_steps=-1)
    with pytest.raises(ValueError, match="dtype"):
        BigClamConfig(dtype="fp16")


def test_ladder_descending_first_accept_wins():
    cfg = BigClamConfig(beta=0.1, ls_steps=3)
    lad = cfg.ladder()
    assert lad[0] == 1.0 and len(lad) == 4
    assert lad == sorted(lad, reverse=True)
    assert lad[1] == pytest.approx(0.1)


def test_json_roundtrip_ignores_unknown_keys():
    cfg = BigClamConfig(k=7, dtype="bf16", tol=1e-5)
    s = cfg.to_json()
    back = BigClamConfig.from_json(s)
    assert back == cfg
    # forward compatibility: unknown keys in a stored config are dropped
    s2 = s.replace('{', '{"future_field": 1,', 1)
    assert BigClamConfig.from_json(s2) == cfg


def test_k_grid_geometric():
    ks = k_grid(1000, 9000, 10)
    assert ks[0] == 1000 and ks[-1] <= 9000
    assert all(b > a for a, b in zip(ks, ks[1:]))
    # reference v4 repl output regression lives in test_engine; here just
    # the generic contract: ~geometric spacing by (1 + 1/k_div)-ish steps
    assert len(ks) > 5
