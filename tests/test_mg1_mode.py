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
