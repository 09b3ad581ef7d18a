"""Unit tests for the queueing analyzer (mirrors the reference's
pkg/analyzer test strategy: constructor validation, prefill/decode formulas,
service-rate construction, Analyze/Size expectations, EffectiveConcurrency,
binary search), plus a differential test of the log-space chain against the
Go-style forward-recurrence oracle."""
import math

import numpy as np
import pytest

from inferno_amd.analyzer import (
    EPSILON,
    AnalyzerError,
    Configuration,
    DecodeParms,
    MM1K,
    MG1K,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    StateDependentChain,
    TargetPerf,
    binary_search,
    build_service_rates,
    effective_concurrency,
    within_tolerance,
)
from tests.oracle import chain_stats_recurrence


def make_parms(alpha=20.58, beta=0.41, gamma=5.2, delta=0.1):
    return ServiceParms(prefill=PrefillParms(gamma, delta), decode=DecodeParms(alpha, beta))


def make_analyzer(N=8, in_tok=128, out_tok=64, **kw):
    cfg = Configuration(max_batch_size=N, max_queue_size=10 * N, service_parms=make_parms(**kw))
    return QueueAnalyzer(cfg, RequestSize(in_tok, out_tok))


class TestFormulas:
    def test_prefill_time(self):
        p = PrefillParms(gamma=5.2, delta=0.1)
        assert p.prefill_time(0, 4.0) == 0.0
        assert p.prefill_time(100, 2.0) == pytest.approx(5.2 + 0.1 * 100 * 2.0, rel=1e-6)

    def test_decode_time(self):
        d = DecodeParms(alpha=20.58, beta=0.41)
        assert d.decode_time(4.0) == pytest.approx(20.58 + 0.41 * 4.0, rel=1e-6)

    def test_service_rates_basic(self):
        cfg = Configuration(4, 40, make_parms())
        serv = build_service_rates(cfg, RequestSize(128, 64))
        assert len(serv) == 4
        for n in range(1, 5):
            prefill = np.float32(5.2) + np.float32(0.1) * np.float32(128) * np.float32(n)
            decode = np.float32(63) * (np.float32(20.58) + np.float32(0.41) * np.float32(n))
            assert serv[n - 1] == pytest.approx(n / (prefill + decode), rel=1e-6)
        # service rate should increase with batch in this regime
        assert np.all(np.diff(serv) > 0)

    def test_service_rates_decode_only_single_token(self):
        # inTok=0 & outTok=1 -> one decode (queueanalyzer.go:105-108)
        cfg = Configuration(2, 20, make_parms())
        serv = build_service_rates(cfg, RequestSize(0, 1))
        for n in (1, 2):
            decode = np.float32(20.58) + np.float32(0.41) * np.float32(n)
            assert serv[n - 1] == pytest.approx(n / decode, rel=1e-6)

    def test_service_rates_prefill_only(self):
        # outTok=1, inTok>0 -> no decode component
        cfg = Configuration(2, 20, make_parms())
        serv = build_service_rates(cfg, RequestSize(100, 1))
        for n in (1, 2):
            prefill = np.float32(5.2) + np.float32(0.1) * np.float32(100) * np.float32(n)
            assert serv[n - 1] == pytest.approx(n / prefill, rel=1e-6)

    def test_effective_concurrency_inverts_service_time(self):
        parms = make_parms()
        req = RequestSize(128, 64)
        for n in (1.0, 2.5, 7.9):
            serv_time = (5.2 + 0.1 * 128 * n) + 63 * (20.58 + 0.41 * n)
            got = effective_concurrency(serv_time, parms, req, 8)
            assert got == pytest.approx(n, rel=1e-6)

    def test_effective_concurrency_clamped(self):
        parms = make_parms()
        req = RequestSize(128, 64)
        assert effective_concurrency(0.0, parms, req, 8) == 0.0
        assert effective_concurrency(1e12, parms, req, 8) == 8.0


class TestValidation:
    def test_bad_config(self):
        with pytest.raises(AnalyzerError):
            QueueAnalyzer(Configuration(0, 10, make_parms()), RequestSize(10, 10))
        with pytest.raises(AnalyzerError):
            QueueAnalyzer(Configuration(4, -1, make_parms()), RequestSize(10, 10))

    def test_bad_request_size(self):
        with pytest.raises(AnalyzerError):
            QueueAnalyzer(Configuration(4, 40, make_parms()), RequestSize(-1, 10))
        with pytest.raises(AnalyzerError):
            QueueAnalyzer(Configuration(4, 40, make_parms()), RequestSize(10, 0))

    def test_analyze_rejects_bad_rates(self):
        qa = make_analyzer()
        with pytest.raises(AnalyzerError):
            qa.analyze(0.0)
        with pytest.raises(AnalyzerError):
            qa.analyze(-1.0)
        with pytest.raises(AnalyzerError):
            qa.analyze(qa.rate_max * 1.01)

    def test_bad_targets(self):
        qa = make_analyzer()
        with pytest.raises(AnalyzerError):
            qa.size(TargetPerf(target_ttft=-1))


class TestChainDifferential:
    """Log-space closed form vs Go-style float64 forward recurrence."""

    @pytest.mark.parametrize("seed", range(8))
    def test_random_grids(self, seed):
        rng = np.random.default_rng(seed)
        N = int(rng.integers(1, 300))
        K = 11 * N
        alpha = rng.uniform(1, 50)
        beta = rng.uniform(0.01, 2)
        gamma = rng.uniform(0.1, 20)
        delta = rng.uniform(0.001, 0.5)
        in_tok = int(rng.integers(0, 2000))
        out_tok = int(rng.integers(1, 1000))
        cfg = Configuration(N, 10 * N, make_parms(alpha, beta, gamma, delta))
        serv = build_service_rates(cfg, RequestSize(in_tok, out_tok))
        chain = StateDependentChain(K, serv)
        lam_max = float(serv[-1]) * (1 - EPSILON)
        lam_min = float(serv[0]) * EPSILON
        for frac in (0.0, 0.1, 0.5, 0.9, 1.0):
            lam = lam_min + frac * (lam_max - lam_min)
            got = chain.solve(lam)
            want = chain_stats_recurrence(K, serv, lam)
            assert got.p0 == pytest.approx(want["p0"], rel=1e-8, abs=1e-300)
            assert got.pK == pytest.approx(want["pK"], rel=1e-6, abs=1e-300)
            assert got.throughput == pytest.approx(want["throughput"], rel=1e-8)
            assert got.avg_num_in_system == pytest.approx(want["avg_num_in_system"], rel=1e-7)
            assert got.avg_num_in_servers == pytest.approx(
                want["avg_num_in_servers"], rel=1e-7, abs=1e-9
            )
            assert got.avg_wait_time == pytest.approx(want["avg_wait_time"], rel=1e-5, abs=1e-4)

    def test_single_batch_state(self):
        # N=1 degenerates to classic M/M/1/K: compare against closed form
        serv = np.array([0.5], dtype=np.float32)
        K = 11
        chain = StateDependentChain(K, serv)
        mm1k = MM1K(K)
        for lam in (0.01, 0.2, 0.4, 0.499):
            got = chain.solve(lam)
            want = mm1k.solve(lam, 0.5)
            assert got.p0 == pytest.approx(want.p0, rel=1e-9)
            assert got.pK == pytest.approx(want.pK, rel=1e-9)
            assert got.throughput == pytest.approx(want.throughput, rel=1e-9)
            assert got.avg_num_in_system == pytest.approx(want.avg_num_in_system, rel=1e-9)


class TestAnalyze:
    def test_monotone_in_rate(self):
        qa = make_analyzer(N=16)
        rates = np.linspace(qa.rate_min, qa.rate_max, 20)
        waits = [qa.analyze(float(r)).avg_wait_time for r in rates]
        itls = [qa.analyze(float(r)).avg_token_time for r in rates]
        assert all(b >= a - 1e-9 for a, b in zip(waits, waits[1:]))
        assert all(b >= a - 1e-9 for a, b in zip(itls, itls[1:]))

    def test_low_rate_limits(self):
        qa = make_analyzer(N=8)
        m = qa.analyze(qa.rate_min)
        # nearly idle: throughput ~ rate, rho ~ 0, ITL ~ alpha + beta*eff with small eff
        assert m.throughput == pytest.approx(qa.rate_min, rel=1e-3)
        assert m.rho < 0.2
        assert m.avg_token_time >= 20.58 - 1e-3

    def test_rho_clamped(self):
        qa = make_analyzer(N=4)
        m = qa.analyze(qa.rate_max)
        assert 0.0 <= m.rho <= 1.0


class TestSize:
    def test_loose_targets_hit_max_rate(self):
        qa = make_analyzer(N=8)
        tr, metrics, achieved = qa.size(TargetPerf(target_ttft=1e9, target_itl=1e9))
        assert tr.rate_target_ttft == pytest.approx(qa.rate_max, rel=1e-6)
        assert tr.rate_target_itl == pytest.approx(qa.rate_max, rel=1e-6)
        assert metrics.throughput <= qa.rate_max

    def test_itl_target_achieved(self):
        qa = make_analyzer(N=8)
        target_itl = 22.0  # between alpha=20.58 and alpha+beta*8=23.86
        tr, metrics, achieved = qa.size(TargetPerf(target_itl=target_itl))
        assert achieved.target_itl <= target_itl * (1 + 1e-3)
        # sized rate reproduces the target when re-evaluated
        assert qa._eval_itl(tr.rate_target_itl / 1000.0) == pytest.approx(target_itl, rel=1e-3)

    def test_ttft_target_achieved(self):
        qa = make_analyzer(N=8)
        lo = qa._eval_ttft(qa.rate_min / 1000.0)
        hi = qa._eval_ttft(qa.rate_max / 1000.0)
        target = 0.5 * (lo + hi)
        tr, metrics, achieved = qa.size(TargetPerf(target_ttft=target))
        assert qa._eval_ttft(tr.rate_target_ttft / 1000.0) == pytest.approx(target, rel=1e-3)

    def test_infeasible_target_below_region(self):
        qa = make_analyzer(N=8)
        # ITL below alpha can never be met
        with pytest.raises(AnalyzerError):
            qa.size(TargetPerf(target_itl=1.0))

    def test_tps_target_stability_margin(self):
        qa = make_analyzer(N=8)
        tr, _, _ = qa.size(TargetPerf(target_tps=100.0))
        assert tr.rate_target_tps == pytest.approx(qa.rate_max * 0.9, rel=1e-6)

    def test_zero_targets_mean_unconstrained(self):
        qa = make_analyzer(N=8)
        tr, _, _ = qa.size(TargetPerf())
        assert tr.rate_target_ttft == pytest.approx(qa.rate_max, rel=1e-6)
        assert tr.rate_target_itl == pytest.approx(qa.rate_max, rel=1e-6)
        assert tr.rate_target_tps == pytest.approx(qa.rate_max, rel=1e-6)


class TestBinarySearch:
    def test_increasing(self):
        x, ind = binary_search(0.0, 10.0, 25.0, lambda x: x * x)
        assert ind == 0
        assert x == pytest.approx(5.0, rel=1e-5)

    def test_decreasing(self):
        x, ind = binary_search(1.0, 10.0, 0.5, lambda x: 1.0 / x)
        assert ind == 0
        assert x == pytest.approx(2.0, rel=1e-5)

    def test_below_region(self):
        x, ind = binary_search(1.0, 10.0, 0.5, lambda x: x)
        assert ind == -1 and x == 1.0

    def test_above_region(self):
        x, ind = binary_search(1.0, 10.0, 20.0, lambda x: x)
        assert ind == +1 and x == 10.0

    def test_boundary_within_tolerance(self):
        x, ind = binary_search(1.0, 10.0, 1.0, lambda x: x)
        assert ind == 0 and x == 1.0

    def test_invalid_range(self):
        with pytest.raises(AnalyzerError):
            binary_search(10.0, 1.0, 5.0, lambda x: x)

    def test_within_tolerance(self):
        assert within_tolerance(1.0, 1.0, 0.0)
        assert within_tolerance(1.0000001, 1.0, 1e-6)
        assert not within_tolerance(1.1, 1.0, 1e-6)
        assert not within_tolerance(0.1, 0.0, 1e-6)


class TestMM1K:
    def test_textbook_values(self):
        m = MM1K(3)
        st = m.solve(1.0, 2.0)  # rho = 0.5, K=3
        # p0 = (1-r)/(1-r^4) = 0.5/0.9375
        assert st.p0 == pytest.approx(0.5 / 0.9375, rel=1e-9)
        assert st.pK == pytest.approx(st.p0 * 0.5**3, rel=1e-9)
        # E[n] = r/(1-r) - 4 r^4/(1-r^4)
        want_n = 0.5 / 0.5 - 4 * 0.5**4 / (1 - 0.5**4)
        assert st.avg_num_in_system == pytest.approx(want_n, rel=1e-9)

    def test_rho_one(self):
        st = MM1K(4).solve(1.0, 1.0)
        assert st.p0 == pytest.approx(1.0 / 5.0, rel=1e-9)
        assert st.avg_num_in_system == pytest.approx(2.0, rel=1e-9)

    def test_invalid(self):
        assert not MM1K(4).solve(1.0, 0.0).is_valid
        assert not MM1K(4).solve(-1.0, 1.0).is_valid

    def test_mg1k_cv1_matches_mm1k(self):
        a = MM1K(10).solve(0.5, 1.0)
        b = MG1K(10, cv2=1.0).solve(0.5, 1.0)
        assert b.avg_wait_time == pytest.approx(a.avg_wait_time, rel=1e-12)

    def test_mg1k_deterministic_halves_wait(self):
        a = MM1K(10).solve(0.5, 1.0)
        b = MG1K(10, cv2=0.0).solve(0.5, 1.0)
        assert b.avg_wait_time == pytest.approx(0.5 * a.avg_wait_time, rel=1e-12)


class TestFlatTailBranch:
    """The r = lam/s(N) == 1 geometric-tail special case (flat tail): the
    closed form switches to the arithmetic branch; it must agree with the
    reference-style forward recurrence exactly like the generic branch."""

    def test_flat_tail_matches_oracle(self):
        import numpy as np

        from inferno_amd.analyzer.queue import StateDependentChain
        from tests.oracle import chain_stats_recurrence

        serv = np.linspace(0.05, 0.4, 16).astype(np.float32)
        K = 11 * len(serv)
        chain = StateDependentChain(K, serv)
        lam = float(serv[-1])  # exactly s(N): r == 1 -> flat tail
        got = chain.solve(lam)
        want = chain_stats_recurrence(K, serv, lam)
        assert got.throughput == pytest.approx(want["throughput"], rel=1e-9)
        assert got.avg_num_in_system == pytest.approx(
            want["avg_num_in_system"], rel=1e-9)
        assert got.pK == pytest.approx(want["pK"], rel=1e-9)
        assert got.p0 == pytest.approx(want["p0"], rel=1e-9)

    def test_near_flat_tail_continuity(self):
        """Values just below/above r=1 bracket the flat-tail value (the
        branch switch introduces no discontinuity)."""
        import numpy as np

        from inferno_amd.analyzer.queue import StateDependentChain

        serv = np.linspace(0.05, 0.4, 16).astype(np.float32)
        chain = StateDependentChain(11 * len(serv), serv)
        sN = float(serv[-1])
        at = chain.solve(sN).avg_num_in_system
        below = chain.solve(sN * (1 - 1e-9)).avg_num_in_system
        above = chain.solve(sN * (1 + 1e-9)).avg_num_in_system
        assert below <= at <= above or above <= at <= below
        assert at == pytest.approx(below, rel=1e-5)
        assert at == pytest.approx(above, rel=1e-5)

    def test_zero_queue_K_equals_N(self):
        """K == N (no queue states): tail sums vanish; compare to oracle."""
        import numpy as np

        from inferno_amd.analyzer.queue import StateDependentChain
        from tests.oracle import chain_stats_recurrence

        serv = np.linspace(0.1, 0.8, 8).astype(np.float32)
        chain = StateDependentChain(len(serv), serv)
        lam = 0.3
        got = chain.solve(lam)
        want = chain_stats_recurrence(len(serv), serv, lam)
        assert got.throughput == pytest.approx(want["throughput"], rel=1e-9)
        assert got.pK == pytest.approx(want["pK"], rel=1e-9)

    def test_single_state_chain(self):
        """N=1, K=11: pure M/M/1/K degenerate chain vs oracle."""
        import numpy as np

        from inferno_amd.analyzer.queue import StateDependentChain
        from tests.oracle import chain_stats_recurrence

        serv = np.array([0.25], dtype=np.float32)
        chain = StateDependentChain(11, serv)
        for lam in (0.01, 0.2, 0.249):
            got = chain.solve(lam)
            want = chain_stats_recurrence(11, serv, lam)
            assert got.avg_num_in_system == pytest.approx(
                want["avg_num_in_system"], rel=1e-9)
