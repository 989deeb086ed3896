"""Header filter tests — mirror reference pkg/headers/filter_test.go scenarios
(filtering on / disabled / forward-all / case-sensitivity / block precedence)."""

from ggrmcp_amd.headers import HeaderFilter


def test_default_allow_list():
    f = HeaderFilter()
    headers = {
        "Authorization": "Bearer tok",
        "X-Trace-Id": "t1",
        "Cookie": "secret",
        "X-Custom": "nope",
        "User-Agent": "ua",
    }
    out = f.filter_headers(headers)
    assert out == {"Authorization": "Bearer tok", "X-Trace-Id": "t1", "User-Agent": "ua"}


def test_disabled_returns_empty():
    f = HeaderFilter(enabled=False)
    assert f.filter_headers({"Authorization": "x"}) == {}
    assert not f.should_forward("Authorization")


def test_forward_all_still_blocks():
    f = HeaderFilter(forward_all=True)
    out = f.filter_headers({"X-Anything": "1", "Cookie": "no", "Mcp-Session-Id": "s"})
    assert out == {"X-Anything": "1"}


def test_blocked_takes_precedence_over_allowed():
    f = HeaderFilter(allowed=["cookie"], blocked=["cookie"])
    assert not f.should_forward("Cookie")


def test_case_sensitive_mode():
    f = HeaderFilter(allowed=["Authorization"], blocked=[], case_insensitive=False)
    assert f.should_forward("Authorization")
    assert not f.should_forward("authorization")


def test_case_insensitive_default():
    f = HeaderFilter()
    assert f.should_forward("AUTHORIZATION")
    assert not f.should_forward("COOKIE")
