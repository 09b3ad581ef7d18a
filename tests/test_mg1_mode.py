"""M/G/1 analyzer mode (BASELINE config 4's evaluator): closed-form
evaluator behavior, system-level wiring, CPU parity between the direct
evaluator and create_allocation."""
import pytest

from inferno_amd.analyzer import (
    Configuration,
    DecodeParms,
    MM1K,
    PrefillParms,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from inferno_amd.analyzer.mg1 import MG1QueueEvaluator
from inferno_amd.core import System, create_allocation
from inferno_amd.engine import SweepEngine
from tests.fixtures import make_spec


def make_eval(N=8, in_tok=128, out_tok=64, cv2=1.0, alpha=20.58, beta=0.41):
    cfg = Configuration(
        N, 10 * N, ServiceParms(PrefillParms(5.2, 0.1), DecodeParms(alpha, beta))
    )
    return MG1QueueEvaluator(cfg, RequestSize(in_tok, out_tok), cv2=cv2)


class TestMG1Evaluator:
    def test_wait_matches_mm1k_closed_form_at_cv1(self):
        ev = make_eval(cv2=1.0)
        lam = ev.mu * 0.5
        m = MM1K(ev.K).solve(lam, ev.mu)
        p = ev._point(lam)
        assert p.wait == pytest.approx(m.avg_wait_time, rel=1e-9)
        assert p.throughput == pytest.approx(m.throughput, rel=1e-12)

    def test_cv2_scales_wait(self):
        lam = make_eval().mu * 0.7
        w1 = make_eval(cv2=1.0)._point(lam).wait
        w0 = make_eval(cv2=0.0)._point(lam).wait
        w3 = make_eval(cv2=3.0)._point(lam).wait
        assert w0 == pytest.approx(0.5 * w1, rel=1e-9)
        assert w3 == pytest.approx(2.0 * w1, rel=1e-9)

    def test_monotone_evals(self):
        ev = make_eval(N=16)
        lams = [ev.rate_min / 1000 * (1 + i * 200) for i in range(5)]
        lams = [min(l, ev.rate_max / 1000 * 0.999) for l in lams]
        ttfts = [ev._eval_ttft(l) for l in lams]
        itls = [ev._eval_itl(l) for l in lams]
        assert all(b >= a - 1e-12 for a, b in zip(ttfts, ttfts[1:]))
        assert all(b >= a - 1e-12 for a, b in zip(itls, itls[1:]))

    def test_size_itl_target(self):
        ev = make_eval(N=8)
        # ITL = alpha + beta*rho*N: target halfway up the range
        target = 20.58 + 0.41 * 4.0
        tr, metrics, achieved = ev.size(TargetPerf(target_itl=target))
        assert achieved.target_itl <= target * (1 + 1e-3)
        # rho*N = 4 -> lam = mu/2
        assert tr.rate_target_itl == pytest.approx(ev.mu * 0.5 * 1000.0, rel=1e-3)

    def test_infeasible_below_region(self):
        ev = make_eval(N=8)
        with pytest.raises(Exception):
            ev.size(TargetPerf(target_itl=1.0))


class TestSystemWiring:
    def test_analyzer_mode_from_spec(self):
        spec = make_spec(n_servers=2, seed=91)
        spec.optimizer.analyzer = "mg1"
        spec.optimizer.analyzerCV2 = 2.0
        system, _ = System.from_spec(spec)
        assert system.analyzer_mode == "mg1"
        assert system.analyzer_cv2 == 2.0

    def test_mg1_allocation_differs_from_chain(self):
        spec_a = make_spec(n_servers=4, seed=92)
        spec_b = make_spec(n_servers=4, seed=92)
        spec_b.optimizer.analyzer = "mg1"
        sys_a, _ = System.from_spec(spec_a)
        sys_b, _ = System.from_spec(spec_b)
        a = create_allocation(sys_a, "srv-0:ns", "MI355X")
        b = create_allocation(sys_b, "srv-0:ns", "MI355X")
        assert a is not None and b is not None
        # both feasible and sane; the evaluators generally disagree on sizing
        assert b.num_replicas >= 1
        assert 0.0 <= b.rho <= 1.0

    def test_engine_solves_in_mg1_mode(self):
        spec = make_spec(n_servers=6, seed=93)
        spec.optimizer.analyzer = "mg1"
        system, opt = System.from_spec(spec)
        SweepEngine(backend="cpu").solve(system, opt)
        n_alloc = sum(1 for s in system.servers.values() if s.allocation is not None)
        assert n_alloc == 6

    def test_json_roundtrip_keeps_analyzer(self):
        from inferno_amd.config import system_spec_from_json, system_spec_to_json

        spec = make_spec(n_servers=1, seed=94)
        spec.optimizer.analyzer = "mg1"
        spec.optimizer.analyzerCV2 = 1.5
        back = system_spec_from_json(system_spec_to_json(spec))
        assert back.optimizer.analyzer == "mg1"
        assert back.optimizer.analyzerCV2 == 1.5


class TestDegenerateParms:
    def test_zero_parms_infeasible(self):
        from inferno_amd.analyzer import AnalyzerError, Configuration, DecodeParms, PrefillParms, QueueAnalyzer, RequestSize, ServiceParms
        import pytest as _pt

        cfg = Configuration(4, 40, ServiceParms(PrefillParms(0.0, 0.0), DecodeParms(0.0, 0.0)))
        with _pt.raises(AnalyzerError):
            QueueAnalyzer(cfg, RequestSize(10, 10))

    def test_zero_parm_allocation_is_none(self):
        spec = make_spec(n_servers=1, seed=95)
        for m in spec.models:
            m.decodeParms.alpha = 0.0
            m.decodeParms.beta = 0.0
            m.prefillParms.gamma = 0.0
            m.prefillParms.delta = 0.0
        system, _ = System.from_spec(spec)
        assert create_allocation(system, "srv-0:ns", "MI355X") is None

    def test_negative_beta_infeasible(self):
        from inferno_amd.analyzer import AnalyzerError, Configuration, DecodeParms, PrefillParms, QueueAnalyzer, RequestSize, ServiceParms
        import pytest as _pt

        # decode time goes negative at high batch -> negative service rate
        cfg = Configuration(64, 640, ServiceParms(PrefillParms(1.0, 0.001), DecodeParms(2.0, -0.5)))
        with _pt.raises(AnalyzerError):
            QueueAnalyzer(cfg, RequestSize(10, 10))


class TestTextbookVectors:
    """External oracle for the M/G/1/K evaluator:
    classical finite-capacity queueing results (Gross & Harris, 'Fundamentals
    of Queueing Theory', M/M/1/K section; Pollaczek-Khinchine mean-value
    formula), computed analytically with exact fractions — independent of the
    implementation's own algebra."""

    def test_mm1k_gross_harris_vector(self):
        # M/M/1/3, lambda=3, mu=4 (rho=3/4):
        #   p0 = 64/175, pK = 27/175, L = 201/175,
        #   X = 444/175, W = 201/444, Wq = 15/74
        from inferno_amd.analyzer.mm1k import MM1K

        st = MM1K(3).solve(3.0, 4.0)
        assert st.is_valid
        assert st.p0 == pytest.approx(64 / 175, rel=1e-12)
        assert st.pK == pytest.approx(27 / 175, rel=1e-12)
        assert st.avg_num_in_system == pytest.approx(201 / 175, rel=1e-9)
        assert st.throughput == pytest.approx(444 / 175, rel=1e-12)
        assert st.avg_resp_time == pytest.approx(201 / 444, rel=1e-9)
        assert st.avg_wait_time == pytest.approx(15 / 74, rel=1e-9)

    def test_mm11_erlang_loss(self):
        # M/M/1/1 (pure loss system) at rho=1/2: p0=2/3, p1=1/3,
        # X = lambda*(1-p1) = 1/3, L = 1/3, W = 1/mu = 1, Wq = 0.
        # (rho=1 with K=1 is OUTSIDE the reference's validity region
        # rho < rhoMax=K, queuemodel.go:27-37 — also asserted here.)
        from inferno_amd.analyzer.mm1k import MM1K

        st = MM1K(1).solve(0.5, 1.0)
        assert st.is_valid
        assert st.p0 == pytest.approx(2 / 3, rel=1e-12)
        assert st.pK == pytest.approx(1 / 3, rel=1e-12)
        assert st.throughput == pytest.approx(1 / 3, rel=1e-12)
        assert st.avg_num_in_system == pytest.approx(1 / 3, rel=1e-9)
        assert st.avg_resp_time == pytest.approx(1.0, rel=1e-9)
        assert st.avg_wait_time == pytest.approx(0.0, abs=1e-12)
        assert MM1K(1).solve(1.0, 1.0).is_valid is False

    def test_pollaczek_khinchine_limit(self):
        # Large K -> unbounded M/G/1; PK mean wait Wq = rho/(1-rho) *
        # (1+cv2)/2 * (1/mu). rho=1/2, mu=1:
        #   cv2=1 (exponential): Wq = 1.0 (M/M/1)
        #   cv2=0 (deterministic): Wq = 0.5
        #   cv2=4 (heavy-tailed):  Wq = 2.5
        from inferno_amd.analyzer.mm1k import MG1K

        for cv2, want in ((1.0, 1.0), (0.0, 0.5), (4.0, 2.5)):
            st = MG1K(10000, cv2=cv2).solve(0.5, 1.0)
            assert st.is_valid
            assert st.avg_wait_time == pytest.approx(want, rel=1e-6), f"cv2={cv2}"

    def test_mm1k_utilization_one(self):
        # rho == 1 degenerate case: p_i = 1/(K+1) uniformly; K=4:
        # p0=pK=1/5, X = lambda*(1-pK) = 4/5, L = K/2 = 2
        from inferno_amd.analyzer.mm1k import MM1K

        st = MM1K(4).solve(1.0, 1.0)
        assert st.p0 == pytest.approx(0.2, rel=1e-9)
        assert st.pK == pytest.approx(0.2, rel=1e-9)
        assert st.throughput == pytest.approx(0.8, rel=1e-9)
        assert st.avg_num_in_system == pytest.approx(2.0, rel=1e-9)
