"""Session manager tests — mirror reference pkg/session/manager.go behavior."""

from ggrmcp_amd.session import SessionManager


def test_create_and_reuse():
    m = SessionManager()
    s1 = m.get_or_create(None, {"authorization": "a"})
    assert len(s1.id) == 32  # 16 bytes hex
    s2 = m.get_or_create(s1.id)
    assert s2 is s1


def test_unknown_id_creates_new_with_that_id():
    m = SessionManager()
    s = m.get_or_create("client-chosen-id")
    assert s.id == "client-chosen-id"


def test_ttl_expiry():
    m = SessionManager(ttl_s=0.0)
    s = m.get_or_create(None)
    assert m.get(s.id) is None  # expired immediately


def test_max_sessions_evicts_oldest():
    m = SessionManager(max_sessions=3)
    ids = [m.get_or_create(None).id for _ in range(4)]
    assert m.get(ids[0]) is None
    assert m.get(ids[3]) is not None
    assert len(m.active_ids()) == 3


def test_call_count_and_stats():
    m = SessionManager()
    s = m.get_or_create(None)
    s.increment_call_count()
    s.increment_call_count()
    assert m.stats()["totalCalls"] == 2
    info = m.session_info(s.id)
    assert info["callCount"] == 2


def test_rate_limit_fixed_window():
    m = SessionManager(rate_limit_per_min=2, rate_limit_burst=1)
    s = m.get_or_create(None)
    assert m.check_rate_limit(s)
    assert m.check_rate_limit(s)
    assert m.check_rate_limit(s)
    assert not m.check_rate_limit(s)  # 2+1 exhausted


def test_block_unblock():
    m = SessionManager()
    s = m.get_or_create(None)
    assert m.block(s.id)
    assert m.get(s.id).is_blocked
    assert m.unblock(s.id)
    assert not m.get(s.id).is_blocked
    assert not m.block("missing")


def test_shard_stability():
    m = SessionManager()
    s = m.get_or_create("some-session")
    assert s.shard(8) == m.get_or_create("some-session").shard(8)
    assert s.shard(1) == 0
    assert 0 <= s.shard(8) < 8


def test_shard_distribution():
    m = SessionManager()
    shards = [m.get_or_create(None).shard(8) for _ in range(800)]
    counts = [shards.count(i) for i in range(8)]
    assert all(c > 40 for c in counts)  # roughly uniform


def test_guard_fast_path():
    # one-lock serving fast lane == the 3-call sequence semantically
    m = SessionManager(rate_limit_per_min=2, rate_limit_burst=0)
    ctx, v = m.guard(None, {"user-agent": "x"})
    assert v == 0 and ctx.id and ctx.call_count == 1
    assert ctx.headers == {"user-agent": "x"}
    # existing session: same ctx, counter advances
    ctx2, v = m.guard(ctx.id)
    assert v == 0 and ctx2 is ctx and ctx.call_count == 2
    # window exhausted -> verdict 2, call_count NOT bumped
    _, v = m.guard(ctx.id)
    assert v == 2 and ctx.call_count == 2
    # blocked wins over rate limit
    m.block(ctx.id)
    _, v = m.guard(ctx.id)
    assert v == 1
    # rate_limit=False skips the window but still counts the call
    m.unblock(ctx.id)
    _, v = m.guard(ctx.id, rate_limit=False)
    assert v == 0 and ctx.call_count == 3
    # expired/unknown id -> fresh session under the same id semantics
    ctx3, v = m.guard("brand-new-id")
    assert v == 0 and ctx3.id == "brand-new-id" and ctx3.call_count == 1
