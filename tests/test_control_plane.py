"""C++ control-plane core tests (hypha_amd._core) with injected clocks —
the deterministic-time analogue of the reference's paused-tokio unit tests
(batch_scheduler.rs:249-483, tracker/*.rs, simulation.rs:71-136,
leases/src/lib.rs)."""

import json

import pytest

core = pytest.importorskip("hypha_amd._core")


class FakeClock:
    def __init__(self):
        self.t = 0.0

    def __call__(self):
        return self.t

    def advance(self, dt):
        self.t += dt


# ---------------------------------------------------------------------------
# json
# ---------------------------------------------------------------------------

def test_json_roundtrip():
    payload = {"a": 1, "b": [1.5, "x", None, True], "c": {"d": "esc\"\n"}}
    out = json.loads(core.json_roundtrip(json.dumps(payload)))
    assert out == payload


# ---------------------------------------------------------------------------
# resources
# ---------------------------------------------------------------------------

def test_resources_partial_order():
    a = core.Resources(1, 2, 3, 4)
    b = core.Resources(2, 2, 3, 4)
    mixed = core.Resources(2, 1, 3, 4)
    assert a.partial_cmp(b) == -1
    assert b.partial_cmp(a) == 1
    assert a.partial_cmp(a) == 0
    assert a.partial_cmp(mixed) is None  # incomparable (lib.rs:123-143)
    assert a.fits_in(b)
    assert not b.fits_in(a)


def test_evaluator_default_weights():
    ev = core.WeightedResourceEvaluator()
    r = core.Resources(1, 1, 10, 100)
    # gpu 25 + cpu 1 + mem 10*0.1 + storage 100*0.01 = 28
    assert ev.weighted_units(r) == pytest.approx(28.0)
    assert ev.score(56.0, r) == pytest.approx(2.0)


# ---------------------------------------------------------------------------
# leases
# ---------------------------------------------------------------------------

def test_ledger_lifecycle():
    clk = FakeClock()
    led = core.Ledger(clock=clk)
    led.insert("l1", "job-a", 10.0)
    assert led.get("l1")[1] == "job-a"
    clk.advance(8)
    assert led.renew("l1", 10.0)  # renew = now + duration
    clk.advance(9)
    assert led.list_expired() == []  # 8 + 10 = 18 > 17
    clk.advance(2)
    assert led.list_expired() == ["l1"]
    assert not led.renew("l1", 10.0)  # expired leases cannot renew
    assert led.drain_expired() == ["l1"]
    assert led.size() == 0


# ---------------------------------------------------------------------------
# slice tracker
# ---------------------------------------------------------------------------

def test_slices_round_robin_and_epoch():
    t = core.SliceTracker("mnist", 4)
    seen = [t.next("w1")[0], t.next("w2")[0], t.next("w1")[0], t.next("w2")[0]]
    assert sorted(seen) == [0, 1, 2, 3]
    assert t.epoch == 0
    # exhausted, equal (unknown) stats -> no steal target beats requester -> rollover
    idx, epoch = t.next("w1")
    assert epoch == 1


def test_slices_cache_steal_from_slowest():
    t = core.SliceTracker("ds", 2)
    t.set_statistic("slow", 1000.0)
    t.set_statistic("fast", 10.0)
    a = t.next("slow")[0]
    b = t.next("slow")[0]
    # fast worker arrives with nothing left: steals slow's latest slice
    idx, epoch = t.next("fast")
    assert epoch == 0  # same epoch: stolen, not rolled over
    assert idx == b
    assert t.owner(idx) == "fast"


def test_slices_remove_worker_reclaims():
    t = core.SliceTracker("ds", 3)
    t.next("w1")
    t.next("w1")
    assert t.available_count() == 1
    t.remove_worker("w1")
    assert t.available_count() == 3


# ---------------------------------------------------------------------------
# progress tracker
# ---------------------------------------------------------------------------

def test_progress_counter_and_rounds():
    p = core.ProgressTracker(100, 2)
    p.on_status(60)
    p.on_status(60)
    assert p.counter == -20
    assert not p.training_finished()
    p.next_round()
    assert p.counter == 100 and p.round == 1
    p.next_round()
    assert p.training_finished()


# ---------------------------------------------------------------------------
# batch scheduler FSM (the reference's scripted multi-worker sequences)
# ---------------------------------------------------------------------------

def test_fsm_single_worker_round():
    clk = FakeClock()
    s = core.BatchScheduler(4, 1, clock=clk)
    s.add_worker("w", 2)
    # first status: no timing stats yet -> simulation can't fit -> continue
    kind, _ = s.handle("w", "status", 2)
    assert kind == "continue"
    clk.advance(0.1)
    # second status: mean known (100ms); remaining counter 0 after this
    kind, counter = s.handle("w", "status", 2)
    assert kind == "schedule-update"
    assert s.worker_state("w") == core.WorkerState.UpdateScheduled
    kind, _ = s.handle("w", "update", 0)
    assert s.worker_state("w") == core.WorkerState.Updating
    # aggregator applied the round
    s.handle("aggregator", "updated", 0)
    kind, _ = s.handle("w", "update-received", 0)
    assert kind == "done"  # 1 round configured
    assert s.worker_state("w") == core.WorkerState.Done
    assert s.finished()


def test_fsm_two_workers_heterogeneous():
    clk = FakeClock()
    s = core.BatchScheduler(12, 2, clock=clk)
    s.add_worker("fast", 2)
    s.add_worker("slow", 1)
    # drive interleaved statuses (fast = 50ms/batch, slow = 200ms/batch)
    # until each worker receives its personal ScheduleUpdate counter
    sched = {}
    for i in range(40):
        clk.advance(0.05)
        if "fast" not in sched:
            k, c = s.handle("fast", "status", 2)
            if k == "schedule-update":
                sched["fast"] = c
        if i % 4 == 3 and "slow" not in sched:
            k, c = s.handle("slow", "status", 1)
            if k == "schedule-update":
                sched["slow"] = c
        if len(sched) == 2:
            break
    assert set(sched) == {"fast", "slow"}
    # heterogeneous shares: the faster worker is never assigned less work
    assert sched["fast"] >= sched["slow"]
    s.handle("fast", "update", 0)
    assert s.worker_state("fast") == core.WorkerState.Updating
    s.handle("slow", "update", 0)
    s.handle("agg", "updated", 0)
    assert s.handle("fast", "update-received", 0)[0] == "continue"
    assert s.handle("slow", "update-received", 0)[0] == "continue"
    assert s.round == 1
    assert s.worker_state("fast") == core.WorkerState.Training


# ---------------------------------------------------------------------------
# auction
# ---------------------------------------------------------------------------

def test_aggregator_ranks_by_score_and_caps_price():
    agg = core.GreedyOfferAggregator(2, core.PriceRange(bid=1.0, max=5.0), 100.0)
    r = core.Resources(1, 8, 32, 100)
    assert not agg.add(core.WorkerOffer("l3", "rq", "w-overpriced", 9.0, r, 50.0), 0.0)  # > max
    assert not agg.add(core.WorkerOffer("l1", "rq", "w-expensive", 4.0, r, 50.0), 0.0)
    full = agg.add(core.WorkerOffer("l2", "rq", "w-cheap", 1.0, r, 50.0), 0.0)
    assert full  # capacity 2 reached -> early return (allocator.rs:316-419)
    best = agg.finalize()
    assert [o.worker for o in best] == ["w-cheap", "w-expensive"]  # score-ranked


def test_aggregator_dedups_per_worker_and_shrinks_deadline():
    agg = core.GreedyOfferAggregator(3, core.PriceRange(bid=1.0, max=5.0), 100.0)
    r = core.Resources(1, 1, 1, 1)
    agg.add(core.WorkerOffer("l1", "rq", "w1", 1.0, r, 7.0), 0.0)
    assert agg.deadline == 7.0  # shrank to earliest offer expiry
    agg.add(core.WorkerOffer("l2", "rq", "w1", 0.5, r, 9.0), 0.0)  # dup ignored
    assert len(agg.finalize()) == 1


def test_arbiter_select_requests():
    policy = core.OfferPolicy(1.0, 0.5, ["diloco-transformer"])
    avail = core.Resources(2, 16, 64, 200)
    ads = [
        core.WorkerRequest("r1", "s", core.Resources(1, 8, 32, 100), ["diloco-transformer"], 2.0),
        core.WorkerRequest("r2", "s", core.Resources(1, 8, 32, 100), ["diloco-transformer"], 0.1),  # below floor
        core.WorkerRequest("r3", "s", core.Resources(8, 8, 32, 100), ["diloco-transformer"], 5.0),  # doesn't fit
        core.WorkerRequest("r4", "s", core.Resources(1, 8, 32, 100), ["exotic-executor"], 3.0),  # unsupported
        core.WorkerRequest("r5", "s", core.Resources(1, 8, 32, 100), ["diloco-transformer"], 1.0),
    ]
    picked = core.select_requests(ads, policy, avail)
    assert [d.request.id for d in picked] == ["r1", "r5"]  # best-paying first, greedy fit


def test_static_resource_manager():
    m = core.StaticResourceManager(core.Resources(2, 8, 32, 100))
    assert m.reserve(core.Resources(1, 4, 16, 50))
    assert not m.reserve(core.Resources(2, 1, 1, 1))  # double-checked
    m.release(core.Resources(1, 4, 16, 50))
    assert m.reserve(core.Resources(2, 8, 32, 100))


def test_ps_nesterov_golden_vs_torch(tmp_path):
    """The C++ parameter-server file pipeline (safetensors average + outer
    Nesterov with persistent momentum) must match torch SGD(nesterov=True)
    fed the negated average delta — the reference's golden-value test
    (parameter_server.rs:448-525) applied to OUR production code path."""
    import torch
    from safetensors.torch import load_file, save_file

    torch.manual_seed(5)
    n = 64
    theta = torch.randn(n)
    p_ref = theta.clone().requires_grad_(True)
    opt = torch.optim.SGD([p_ref], lr=0.7, momentum=0.9, nesterov=True)
    mom_path = str(tmp_path / "momentum.safetensors")
    for rnd in range(3):
        deltas = [torch.randn(n) for _ in range(3)]
        files = []
        for i, d in enumerate(deltas):
            fp = str(tmp_path / f"d{rnd}_{i}.safetensors")
            save_file({"w": d}, fp)
            files.append(fp)
        upd_path = str(tmp_path / f"u{rnd}.safetensors")
        core.ps_aggregate_files(files, mom_path, upd_path, 0.7, 0.9)
        update = load_file(upd_path)["w"]
        theta = theta + update  # worker merge: theta <- theta_0 + U
        avg = torch.stack(deltas).mean(0)
        p_ref.grad = -avg
        opt.step()
    torch.testing.assert_close(theta, p_ref.detach(), rtol=1e-4, atol=1e-5)


def test_ps_average_bf16_and_f32(tmp_path):
    import torch
    from safetensors.torch import load_file, save_file

    a = torch.randn(10, dtype=torch.bfloat16)
    b = torch.randn(10, dtype=torch.bfloat16)
    save_file({"t": a}, str(tmp_path / "a.safetensors"))
    save_file({"t": b}, str(tmp_path / "b.safetensors"))
    core.ps_aggregate_files(
        [str(tmp_path / "a.safetensors"), str(tmp_path / "b.safetensors")],
        str(tmp_path / "m.safetensors"), str(tmp_path / "u.safetensors"), 1.0, 0.0)
    u = load_file(str(tmp_path / "u.safetensors"))["t"]
    want = ((a.float() + b.float()) / 2).bfloat16()
    torch.testing.assert_close(u.float(), want.float(), rtol=2e-2, atol=2e-2)


def test_json_parser_fuzz_roundtrip():
    """Property fuzz: the C++ JSON wire codec must round-trip anything the
    Python side can produce (the control plane crosses this boundary on
    every message)."""
    import json as pyjson

    from hypothesis import given, settings, strategies as st

    scalars = st.one_of(
        st.none(), st.booleans(),
        st.integers(min_value=-(2**53) + 1, max_value=2**53 - 1),
        st.floats(allow_nan=False, allow_infinity=False, width=32),
        st.text(max_size=40),
    )
    values = st.recursive(
        scalars,
        lambda children: st.one_of(
            st.lists(children, max_size=4),
            st.dictionaries(st.text(max_size=8), children, max_size=4),
        ),
        max_leaves=20,
    )

    @settings(max_examples=200, deadline=None)
    @given(values)
    def check(v):
        out = pyjson.loads(core.json_roundtrip(pyjson.dumps(v)))
        if isinstance(v, float):
            assert out == pytest.approx(v, rel=1e-6, abs=1e-9)
        else:
            assert out == v

    check()


def test_json_parser_rejects_garbage():
    for bad in ("", "{", "[1,", '{"a"}', "tru", "nul", '"\\u12', "1e999x", "{}{}"):
        with pytest.raises(Exception):
            core.json_roundtrip(bad)


def test_resources_partial_order_properties():
    """Property test (reference worker dev-deps use proptest; resources
    lib.rs:123-143 partial order): reflexive, antisymmetric, consistent
    with fits_in, and incomparable when dimensions disagree in sign."""
    from hypothesis import given, settings, strategies as st

    dim = st.floats(min_value=0, max_value=1e6, allow_nan=False)
    quad = st.tuples(dim, dim, dim, dim)

    @settings(max_examples=200, deadline=None)
    @given(quad, quad)
    def check(a4, b4):
        a = core.Resources(*a4)
        b = core.Resources(*b4)
        assert a.partial_cmp(a) == 0  # reflexive
        c_ab = a.partial_cmp(b)
        c_ba = b.partial_cmp(a)
        if c_ab is None:
            assert c_ba is None  # incomparability is symmetric
        else:
            assert c_ba == -c_ab  # antisymmetry
        # fits_in consistency: a fits in b iff a <= b on every dimension
        fits = all(x <= y for x, y in zip(a4, b4))
        assert a.fits_in(b) == fits
        if fits:
            assert c_ab in (-1, 0)
        # arithmetic round-trip: (a + b) - b == a (within float error)
        rt = (a + b) - b
        for name in ("gpu", "cpu", "memory", "storage"):
            assert abs(getattr(rt, name) - getattr(a, name)) <= 1e-6 * max(
                1.0, getattr(a, name))

    check()


def test_ledger_properties():
    """Property test: leases expire exactly by the injected clock; renew
    extends from NOW (leases lib.rs:103-114), never shortens relative to
    the new timeout; drain removes precisely the expired set."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=100, deadline=None)
    @given(st.lists(st.tuples(st.floats(0.1, 50.0), st.floats(0.0, 100.0)),
                    min_size=1, max_size=20),
           st.floats(0.0, 100.0))
    def check(leases, probe_time):
        now = [0.0]
        led = core.Ledger(clock=lambda: now[0])
        ids = []
        for i, (ttl, _) in enumerate(leases):
            led.insert(f"lease{i}", f"v{i}", ttl)
            ids.append(f"lease{i}")
        now[0] = probe_time
        expired = set(led.list_expired())
        for lease_id, (ttl, _) in zip(ids, leases):
            assert (lease_id in expired) == (probe_time > ttl)
        drained = set(led.drain_expired())
        assert drained == expired
        assert led.size() == len(ids) - len(expired)
        # renew survivors: timeout becomes now + d, so nothing is expired
        # at any probe <= now + min_renewal
        for lease_id in led.ids():
            assert led.renew(lease_id, 5.0)
        now[0] = probe_time + 4.999
        assert led.list_expired() == []

    check()


def test_slice_tracker_properties():
    """Property test: every slice index handed out is valid; each epoch
    serves every slice at least once across workers; removing a worker
    never loses slices (slice.rs:92-114 semantics)."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=50, deadline=None)
    @given(st.integers(1, 12), st.integers(1, 4), st.integers(5, 60))
    def check(num_slices, num_workers, pulls):
        t = core.SliceTracker("ds", num_slices)
        served = {}
        for i in range(pulls):
            peer = f"w{i % num_workers}"
            idx, epoch = t.next(peer)
            assert 0 <= idx < num_slices
            served.setdefault(epoch, set()).add(idx)
        # every completed epoch covered all slices
        max_epoch = max(served)
        for e, s in served.items():
            if e < max_epoch:
                assert s == set(range(num_slices))
        t.remove_worker("w0")
        # after reclaim the tracker still serves valid indices
        idx, _ = t.next("w1")
        assert 0 <= idx < num_slices

    check()


def test_ps_weighted_average(tmp_path):
    """Sample-count-weighted pseudo-gradient mean (the reference leaves this
    as TODO at parameter_server.rs:192-193; here it backs the heterogeneous
    batch path): avg = sum(w_i d_i) / sum(w_i)."""
    import torch
    from safetensors.torch import load_file, save_file

    a = torch.tensor([1.0, 2.0, 3.0])
    b = torch.tensor([4.0, -2.0, 0.0])
    fa, fb = str(tmp_path / "a.safetensors"), str(tmp_path / "b.safetensors")
    save_file({"d": a}, fa)
    save_file({"d": b}, fb)
    out = str(tmp_path / "avg.safetensors")
    core.ps_weighted_average_files([fa, fb], [2.0, 1.0], out)
    got = load_file(out)["d"]
    torch.testing.assert_close(got, (2 * a + b) / 3)
    # equal weights == plain mean
    core.ps_weighted_average_files([fa, fb], [5.0, 5.0], out)
    torch.testing.assert_close(load_file(out)["d"], (a + b) / 2)


def test_simulation_matches_brute_force():
    """Property test for the discrete-event projection (simulation.rs:16-68
    semantics): compared against an independent brute-force reimplementation
    over random worker sets."""
    import heapq

    from hypothesis import given, settings, strategies as st

    def brute(workers, counter, time_cap, update_cap):
        # workers: {peer: (batch, mean_or_None)}
        heap = [((m if m is not None else 1e18), peer)
                for peer, (b, m) in workers.items()]
        heapq.heapify(heap)
        batches = {p: 0 for p in workers}
        time_ms, capped = 0.0, False
        remaining = counter
        while remaining > 0:
            t, peer = heapq.heappop(heap)
            if t > time_cap:
                capped = True
                break
            if batches[peer] + 1 > update_cap:
                capped = True
                break
            batches[peer] += 1
            b, m = workers[peer]
            remaining -= b
            time_ms = t
            heapq.heappush(heap, (t + (m if m is not None else 1e18), peer))
        return time_ms, remaining, batches, capped

    mean = st.one_of(st.none(), st.floats(1.0, 5000.0))
    worker = st.tuples(st.integers(1, 8), mean)

    @settings(max_examples=150, deadline=None)
    @given(st.dictionaries(st.sampled_from(["a", "b", "c", "d"]), worker,
                           min_size=1, max_size=4),
           st.integers(1, 64))
    def check(workers, counter):
        p = core.simulate_project(workers, counter)
        t, rem, batches, capped = brute(workers, counter, 10000.0, 3)
        assert p.remaining == rem
        assert p.capped == capped
        assert p.batches_per_worker == batches
        assert p.time_ms == pytest.approx(t)

    check()


def test_aggregator_properties():
    """Property test for the auction buy side (allocator.rs:276-419
    semantics): the finalized set is exactly the `count` cheapest
    per-weighted-unit acceptable offers, deduped per worker, with the
    deadline shrunk to the earliest offer expiry."""
    from hypothesis import given, settings, strategies as st

    offer = st.tuples(
        st.sampled_from([f"w{i}" for i in range(6)]),   # worker
        st.floats(0.1, 12.0),                            # price
        st.floats(0.5, 4.0),                             # gpu
        st.floats(1.0, 100.0),                           # expiry (abs)
    )

    @settings(max_examples=150, deadline=None)
    @given(st.lists(offer, min_size=1, max_size=12), st.integers(1, 4))
    def check(raw, count):
        price_max = 5.0
        agg = core.GreedyOfferAggregator(count, core.PriceRange(bid=1.0, max=price_max), 1000.0)
        ev = core.WeightedResourceEvaluator()
        accepted, seen, deadline = [], set(), 1000.0
        for i, (w, price, gpu, exp) in enumerate(raw):
            r = core.Resources(gpu, 1, 1, 1)
            agg.add(core.WorkerOffer(f"l{i}", "rq", w, price, r, exp), 0.0)
            if price <= price_max and w not in seen:
                seen.add(w)
                accepted.append((ev.score(price, r), i, w))
                if exp > 0.0:
                    deadline = min(deadline, exp)
        best = agg.finalize()
        model = sorted(accepted, key=lambda t: t[0])[:count]
        assert [o.worker for o in best] == [w for _, _, w in model]
        assert agg.deadline == pytest.approx(deadline)
        # every returned offer is within the price ceiling and unique
        assert len({o.worker for o in best}) == len(best)
        assert all(o.price <= price_max for o in best)

    check()


def test_ps_rejects_malformed_safetensors(tmp_path):
    """A pseudo-gradient file whose data_offsets disagree with shape*itemsize
    (or overrun the data section) is rejected at parse time instead of being
    walked out-of-bounds by the PS math (ADVICE high, safetensors.h)."""
    import json as pyjson
    import struct

    import pytest as _pytest

    def craft(path, dtype, shape, begin, end, payload_len):
        hdr = pyjson.dumps(
            {"t": {"dtype": dtype, "shape": shape, "data_offsets": [begin, end]}}
        ).encode()
        with open(path, "wb") as f:
            f.write(struct.pack("<Q", len(hdr)))
            f.write(hdr)
            f.write(b"\x00" * payload_len)

    mom = str(tmp_path / "m.safetensors")
    upd = str(tmp_path / "u.safetensors")

    # span (8 bytes) != numel*itemsize (40 bytes)
    bad1 = str(tmp_path / "bad1.safetensors")
    craft(bad1, "F32", [10], 0, 8, 8)
    with _pytest.raises(RuntimeError, match="safetensors"):
        core.ps_aggregate_files([bad1], mom, upd, 0.7, 0.9)

    # offsets overrun the actual data section in the file
    bad2 = str(tmp_path / "bad2.safetensors")
    craft(bad2, "F32", [10], 0, 40, 4)
    with _pytest.raises(RuntimeError, match="safetensors"):
        core.ps_aggregate_files([bad2], mom, upd, 0.7, 0.9)

    # negative begin
    bad3 = str(tmp_path / "bad3.safetensors")
    craft(bad3, "F32", [1], -4, 0, 4)
    with _pytest.raises(RuntimeError, match="safetensors"):
        core.ps_aggregate_files([bad3], mom, upd, 0.7, 0.9)

    # a well-formed file still aggregates
    import torch
    from safetensors.torch import save_file

    good = str(tmp_path / "good.safetensors")
    save_file({"t": torch.ones(10)}, good)
    core.ps_aggregate_files([good], mom, upd, 1.0, 0.0)
