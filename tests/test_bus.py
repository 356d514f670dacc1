from colearn_federated_learning_amd.federated.bus import MessageBus, Topics


def test_publish_subscribe():
    bus = MessageBus()
    got = []
    bus.subscribe("a/b", lambda t, p: got.append((t, p)))
    bus.publish("a/b", 1)
    bus.publish("a/c", 2)
    assert got == [("a/b", 1)]


def test_wildcard_plus():
    bus = MessageBus()
    got = []
    bus.subscribe("colearn/role/+", lambda t, p: got.append(t))
    bus.publish("colearn/role/3", {"role": "trainer"})
    bus.publish("colearn/role/3/extra", {})
    assert got == ["colearn/role/3"]


def test_wildcard_hash():
    bus = MessageBus()
    got = []
    bus.subscribe("colearn/#", lambda t, p: got.append(t))
    bus.publish("colearn/availability", {})
    bus.publish("colearn/role/1", {})
    bus.publish("other/x", {})
    assert got == ["colearn/availability", "colearn/role/1"]


def test_retained_delivered_on_subscribe():
    bus = MessageBus()
    bus.publish(Topics.AVAILABILITY, {"client_id": 1}, retain=True)
    got = []
    bus.subscribe(Topics.AVAILABILITY, lambda t, p: got.append(p))
    assert got == [{"client_id": 1}]


def test_collect():
    bus = MessageBus()
    bus.publish("colearn/availability", {"client_id": 0}, retain=True)
    bus.publish("colearn/metrics/0", {"x": 1}, retain=True)
    avail = bus.collect("colearn/availability")
    assert list(avail.values()) == [{"client_id": 0}]


def test_mqtt_wildcard_semantics():
    """'+' is exactly one level; '#' covers all remaining levels and may
    follow '+' levels (real MQTT semantics, not glob)."""
    m = MessageBus._match
    assert m("a/+/b", "a/x/b")
    assert not m("a/+/b", "a/x/y/b")      # '+' must not cross levels
    assert m("a/#", "a/b/c/d")
    assert m("a/+/b/#", "a/x/b/c/d")
    assert not m("a/+/b/#", "a/x/y/b/c")  # '+' crossed a level
    assert m("a/+/#", "a/x/b")
    assert not m("a/+/#", "a")            # '+' needs its level present
    assert m("#", "anything/at/all")


def test_mqtt_match_property():
    """_match vs an independent recursive MQTT matcher on random
    topic/pattern pairs."""
    import itertools
    import random

    def oracle(pp, tp):
        # recursive MQTT semantics: '+' one level, trailing '#' any tail
        if not pp:
            return not tp
        if pp[0] == "#":
            return len(pp) == 1
        if not tp:
            return False
        if pp[0] == "+" or pp[0] == tp[0]:
            return oracle(pp[1:], tp[1:])
        return False

    rng = random.Random(0)
    atoms = ["a", "b", "colearn", "availability", "3"]
    wild = atoms + ["+"]
    for _ in range(2000):
        tlen = rng.randint(1, 4)
        plen = rng.randint(1, 4)
        topic = "/".join(rng.choice(atoms) for _ in range(tlen))
        parts = [rng.choice(wild) for _ in range(plen)]
        if rng.random() < 0.4:
            parts[-1] = "#"
        pattern = "/".join(parts)
        want = oracle(pattern.split("/"), topic.split("/"))
        got = MessageBus._match(pattern, topic)
        assert got == want, (pattern, topic, got, want)
