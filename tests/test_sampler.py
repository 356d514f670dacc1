from colearn_federated_learning_amd.federated.sampler import ClientSampler


def test_plan_deterministic():
    s1 = ClientSampler(seed=1, clients_per_round=4, num_evaluators=2)
    s2 = ClientSampler(seed=1, clients_per_round=4, num_evaluators=2)
    reg = list(range(10))
    for r in range(5):
        p1, p2 = s1.plan(r, reg), s2.plan(r, reg)
        assert p1.trainers == p2.trainers
        assert p1.evaluators == p2.evaluators


def test_roles_disjoint_and_sized():
    s = ClientSampler(seed=3, clients_per_round=5, num_evaluators=3)
    plan = s.plan(0, list(range(12)))
    assert len(plan.trainers) == 5
    assert len(plan.evaluators) == 3
    assert not set(plan.trainers) & set(plan.evaluators)


def test_plan_varies_by_round():
    s = ClientSampler(seed=3, clients_per_round=3)
    reg = list(range(16))
    plans = {tuple(s.plan(r, reg).trainers) for r in range(8)}
    assert len(plans) > 1


def test_all_clients_default():
    s = ClientSampler(seed=0)
    plan = s.plan(0, list(range(6)))
    assert plan.trainers == list(range(6))
    assert plan.evaluators == []


def test_state_roundtrip():
    s = ClientSampler(seed=9, clients_per_round=2, num_evaluators=1)
    d = s.state_dict()
    s2 = ClientSampler(seed=0)
    s2.load_state_dict(d)
    assert s2.plan(3, list(range(8))).trainers == s.plan(3, list(range(8))).trainers
