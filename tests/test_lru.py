from parca_agent_amd.lru import LRU


def test_basic_eviction():
    evicted = []
    c = LRU(max_size=2, on_evict=lambda k, v: evicted.append(k))
    c.put("a", 1)
    c.put("b", 2)
    c.put("c", 3)
    assert c.get("a") is None
    assert c.get("b") == 2
    assert c.get("c") == 3
    assert evicted == ["a"]
    assert c.evictions == 1


def test_lru_ordering():
    c = LRU(max_size=2)
    c.put("a", 1)
    c.put("b", 2)
    c.get("a")  # refresh a
    c.put("c", 3)  # evicts b
    assert c.get("b") is None
    assert c.get("a") == 1


def test_ttl():
    now = [0.0]
    c = LRU(max_size=10, ttl_seconds=5.0, clock=lambda: now[0])
    c.put("a", 1)
    now[0] = 4.9
    assert c.get("a") == 1
    now[0] = 5.1
    assert c.get("a") is None


def test_update_moves_to_end():
    c = LRU(max_size=2)
    c.put("a", 1)
    c.put("b", 2)
    c.put("a", 10)
    c.put("c", 3)  # evicts b, not a
    assert c.get("a") == 10
    assert c.get("b") is None
