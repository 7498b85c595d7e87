"""cubefs_amd.codemode vs the reference registry
(codemode.go:65-94, codemode_test.go)."""
import pytest

from cubefs_amd import codemode as cm


def test_registry_constants():
    assert cm.get_code("EC6P3") == 13
    assert cm.get_name(2) == "EC6P6"
    t = cm.get_tactic("EC6P10L2")
    assert (t.N, t.M, t.L, t.AZCount, t.PutQuorum) == (6, 10, 2, 2, 14)
    assert cm.get_tactic(9) == cm.get_tactic("EC12P4")
    assert cm.is_valid("EC15P12") and not cm.is_valid("ECbogus")
    for code in cm.all_code_modes():
        assert cm.get_tactic(code).is_valid(), code


def test_quorum_invariant():
    """init() assertion (codemode.go:213-219): N + (N+M)/AZ <= PutQuorum <= N+M."""
    for name in ("EC15P12", "EC6P6", "EC12P9", "EC16P20L2", "EC6P10L2"):
        t = cm.get_tactic(name)
        assert t.N + (t.N + t.M) // t.AZCount <= t.PutQuorum <= t.N + t.M, name


def test_ec_layout_by_az():
    """The reference's own comment (codemode.go:152-158) for EC6P10L2."""
    t = cm.get_tactic("EC6P10L2")
    s = t.ec_layout_by_az()
    assert s[0] == [0, 1, 2, 6, 7, 8, 9, 10, 16]
    assert s[1] == [3, 4, 5, 11, 12, 13, 14, 15, 17]
    stripes, n, m = t.all_local_stripe()
    assert (n, m) == (8, 1)
    # LocalStripe by member index (codemode.go:341-358)
    for idx in (0, 1, 2):
        st, _, _ = t.local_stripe(idx)
        assert st == s[0]
    for idx in (3, 11, 17):
        st, _, _ = t.local_stripe(idx)
        assert st == s[1]
    assert t.local_stripe(18) == ([], 0, 0)


def test_global_stripe():
    t = cm.get_tactic("EC6P6")
    idx, n, m = t.global_stripe()
    assert idx == list(range(12)) and (n, m) == (6, 6)
    assert t.all_local_stripe() == ([], 0, 0)


def test_extend():
    """codemode.Extend (codemode.go:399-441): the Azure-LRC(12,2,2) mode
    BASELINE.json names is registered this way."""
    t = cm.Tactic(12, 2, 2, 2, 14, 0, 2048)
    cm.extend(240, "LRC12P2L2", t)
    assert cm.get_tactic(240) == t
    cm.extend(240, "LRC12P2L2", t)  # idempotent
    with pytest.raises(ValueError):
        cm.extend(240, "LRC12P2L2", cm.Tactic(12, 4, 2, 2, 16, 0, 2048))
    with pytest.raises(ValueError):
        cm.extend(100, "TooLow", t)
    with pytest.raises(ValueError):
        cm.extend(241, "BadQuorum", cm.Tactic(12, 2, 2, 2, 1, 0, 2048))


def test_replicate_modes():
    assert cm.get_tactic("Replica3").is_replicate()
    assert not cm.get_tactic("EC6P3").is_replicate()
    assert 100 not in cm.ec_code_modes()
    assert 13 in cm.ec_code_modes()
