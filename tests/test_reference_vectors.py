"""Parity suite: the reference's OWN unit-test vectors, re-expressed against
our analyzer API.

Each case mirrors a numeric expectation or error-semantics case from the
reference's Go test files (pkg/analyzer/queueanalyzer_test.go,
pkg/analyzer/utils_test.go) — the values and expected outcomes come from the
reference's published behavior, the assertions run against our
implementation. Passing this file means a user's mental model built on the
reference transfers unchanged.
"""
import math

import pytest

from inferno_amd.analyzer.queue import (
    AnalyzerError,
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
    binary_search,
    effective_concurrency,
    within_tolerance,
)


def _ref_config():
    return Configuration(
        max_batch_size=8,
        max_queue_size=16,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=10.0, delta=0.001),
            decode=DecodeParms(alpha=1.0, beta=0.01),
        ),
    )


class TestPrefillTime:
    """Ref: queueanalyzer_test.go TestPrefillParms_PrefillTime."""

    @pytest.mark.parametrize(
        "in_tok,batch,expected",
        [
            (0, 4.0, 0.0),       # no input tokens
            (1000, 1.0, 11.0),   # 10.0 + 0.001*1000*1.0
            (2000, 8.0, 26.0),   # 10.0 + 0.001*2000*8.0
            (500, 2.5, 11.25),   # 10.0 + 0.001*500*2.5
        ],
    )
    def test_vectors(self, in_tok, batch, expected):
        p = PrefillParms(gamma=10.0, delta=0.001)
        assert abs(p.prefill_time(in_tok, batch) - expected) <= 1e-6


class TestDecodeTime:
    """Ref: queueanalyzer_test.go TestDecodeParms_DecodeTime."""

    @pytest.mark.parametrize(
        "batch,expected",
        [(1.0, 1.01), (4.0, 1.04), (8.0, 1.08), (2.5, 1.025)],
    )
    def test_vectors(self, batch, expected):
        d = DecodeParms(alpha=1.0, beta=0.01)
        assert abs(d.decode_time(batch) - expected) <= 1e-6


class TestRequestSizeCheck:
    """Ref: queueanalyzer_test.go TestRequestSize_Check + TestNewQueueAnalyzer."""

    @pytest.mark.parametrize(
        "in_tok,out_tok,want_err",
        [
            (100, 10, False),  # valid
            (0, 10, False),    # zero input tokens ok (decode only)
            (100, 1, False),   # minimum output tokens
            (0, 1, False),     # no prefill, one output token
            (-1, 10, True),    # negative input tokens
            (100, 0, True),    # zero output tokens
            (100, -1, True),   # negative output tokens
            (0, 0, True),      # zero input AND output
            (-1, -1, True),    # both negative
            (50, 0, True),     # no decode, no first output token
        ],
    )
    def test_vectors(self, in_tok, out_tok, want_err):
        req = RequestSize(avg_input_tokens=in_tok, avg_output_tokens=out_tok)
        if want_err:
            with pytest.raises(AnalyzerError):
                QueueAnalyzer(_ref_config(), req)
        else:
            QueueAnalyzer(_ref_config(), req)  # must not raise


class TestConfigurationCheck:
    """Ref: queueanalyzer_test.go TestConfiguration_Check."""

    def _cfg(self, batch=8, queue=16, parms="ok"):
        base = _ref_config().service_parms
        if parms == "ok":
            sp = base
        elif parms == "none":
            sp = None
        elif parms == "no_prefill":
            sp = ServiceParms(prefill=None, decode=base.decode)
        elif parms == "no_decode":
            sp = ServiceParms(prefill=base.prefill, decode=None)
        return Configuration(max_batch_size=batch, max_queue_size=queue,
                             service_parms=sp)

    @pytest.mark.parametrize(
        "kwargs,want_err",
        [
            ({}, False),                      # valid configuration
            ({"batch": 0}, True),             # zero max batch size
            ({"batch": -1}, True),            # negative max batch size
            ({"queue": -1}, True),            # negative max queue size
            ({"parms": "none"}, True),        # nil service parameters
            ({"parms": "no_prefill"}, True),  # nil prefill parameters
            ({"parms": "no_decode"}, True),   # nil decode parameters
        ],
    )
    def test_vectors(self, kwargs, want_err):
        cfg = self._cfg(**kwargs)
        req = RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        if want_err:
            with pytest.raises((AnalyzerError, AttributeError, TypeError)):
                QueueAnalyzer(cfg, req)
        else:
            QueueAnalyzer(cfg, req)


class TestRateRange:
    """Ref: queueanalyzer_test.go TestBuildModel — RateRange invariants."""

    def test_rate_range(self):
        qa = QueueAnalyzer(_ref_config(),
                           RequestSize(avg_input_tokens=100, avg_output_tokens=10))
        assert qa.rate_min < qa.rate_max
        assert qa.rate_min > 0
        assert qa.max_batch_size == 8
        assert qa.max_queue_size == 16


class TestAnalyze:
    """Ref: queueanalyzer_test.go TestQueueAnalyzer_Analyze."""

    @pytest.fixture
    def qa(self):
        return QueueAnalyzer(_ref_config(),
                             RequestSize(avg_input_tokens=100, avg_output_tokens=10))

    def test_zero_rate_errors(self, qa):
        with pytest.raises(AnalyzerError):
            qa.analyze(0.0)

    def test_negative_rate_errors(self, qa):
        with pytest.raises(AnalyzerError):
            qa.analyze(-1.0)

    def test_exceeding_rate_errors(self, qa):
        with pytest.raises(AnalyzerError):
            qa.analyze(qa.rate_max * 1.1)

    @pytest.mark.parametrize("frac", ["low", "medium", "high"])
    def test_valid_rates(self, qa, frac):
        rate = {
            "low": qa.rate_min * 0.5,
            "medium": (qa.rate_min + qa.rate_max) * 0.5,
            "high": qa.rate_max * 0.9,
        }[frac]
        m = qa.analyze(rate)
        assert m.throughput >= 0
        assert m.avg_resp_time >= 0
        assert m.avg_wait_time >= 0
        assert m.avg_num_in_serv >= 0
        assert 0.0 <= m.rho <= 1.0
        assert m.avg_prefill_time >= 0
        assert m.avg_token_time >= 0


class TestSize:
    """Ref: queueanalyzer_test.go TestQueueAnalyzer_Size."""

    @pytest.fixture
    def qa(self):
        return QueueAnalyzer(_ref_config(),
                             RequestSize(avg_input_tokens=100, avg_output_tokens=10))

    @pytest.mark.parametrize(
        "ttft,itl,tps,want_err",
        [
            (50.0, 5.0, 100.0, False),  # valid targets
            (0.0, 0.0, 0.0, False),     # zero targets (disabled)
            (-1.0, 5.0, 100.0, True),   # negative TTFT
            (50.0, -1.0, 100.0, True),  # negative ITL
            (50.0, 5.0, -1.0, True),    # negative TPS
        ],
    )
    def test_vectors(self, qa, ttft, itl, tps, want_err):
        targets = TargetPerf(target_ttft=ttft, target_itl=itl, target_tps=tps)
        if want_err:
            with pytest.raises(AnalyzerError):
                qa.size(targets)
            return
        target_rate, metrics, achieved = qa.size(targets)
        assert target_rate.rate_target_ttft >= 0
        assert target_rate.rate_target_itl >= 0
        assert target_rate.rate_target_tps >= 0
        assert achieved.target_ttft >= 0
        assert achieved.target_itl >= 0
        assert achieved.target_tps >= 0
        assert metrics is not None


class TestEffectiveConcurrency:
    """Ref: queueanalyzer_test.go TestEffectiveConcurrency — clamped to [0, N]."""

    @pytest.mark.parametrize("serv_time", [20.0, 50.0, 100.0])
    def test_bounds(self, serv_time):
        cfg = _ref_config()
        req = RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        n = effective_concurrency(serv_time, cfg.service_parms, req, 8)
        assert 0.0 <= n <= 8.0


class TestWithinTolerance:
    """Ref: utils_test.go TestWithinTolerance."""

    @pytest.mark.parametrize(
        "x,value,tol,expected",
        [
            (1.0, 1.0, 0.01, True),     # exact match
            (1.005, 1.0, 0.01, True),   # within tolerance
            (1.02, 1.0, 0.01, False),   # outside tolerance
            (0.1, 0.0, 0.01, False),    # zero value
            (1.0, 1.0, -0.01, True),    # exact match beats negative tolerance
            (0.0, 0.0, 0.01, True),     # both zero
        ],
    )
    def test_vectors(self, x, value, tol, expected):
        assert within_tolerance(x, value, tol) is expected


class TestBinarySearch:
    """Ref: utils_test.go TestBinarySearch / TestBinarySearch_EdgeCases."""

    def test_find_square_root(self):
        x, ind = binary_search(0.0, 10.0, 4.0, lambda x: x * x)
        assert ind == 0
        assert abs(x * x - 4.0) <= 0.1

    def test_linear_in_range(self):
        x, ind = binary_search(1.0, 5.0, 6.0, lambda x: 2 * x)  # f(3)=6
        assert ind == 0
        assert abs(2 * x - 6.0) <= 0.1

    def test_target_below_range(self):
        x, ind = binary_search(2.0, 5.0, 1.0, lambda x: 2 * x)  # below f(2)=4
        assert ind == -1 and x == 2.0  # returns xMin

    def test_target_above_range(self):
        x, ind = binary_search(1.0, 3.0, 10.0, lambda x: 2 * x)  # above f(3)=6
        assert ind == 1 and x == 3.0  # returns xMax

    def test_decreasing_function(self):
        x, ind = binary_search(1.0, 5.0, -3.0, lambda x: -x)  # f(3)=-3
        assert ind == 0
        assert abs(-x - (-3.0)) <= 0.1

    def test_invalid_range_errors(self):
        with pytest.raises(AnalyzerError):
            binary_search(5.0, 1.0, 3.0, lambda x: 2 * x)

    def test_eval_error_propagates(self):
        def bad(x):
            if x > 5.0:
                raise AnalyzerError("x too large")
            return x

        with pytest.raises(AnalyzerError):
            binary_search(4.0, 6.0, 5.0, bad)

    def test_target_at_boundary(self):
        x, ind = binary_search(1.0, 5.0, 2.0, lambda x: 2 * x)  # f(1)=2
        assert ind == 0 and x == 1.0

    def test_constant_function_match(self):
        x, ind = binary_search(1.0, 10.0, 5.0, lambda x: 5.0)
        assert ind == 0  # boundary value matches immediately

    def test_quadratic_precision(self):
        """Ref: utils_test.go:625-645 — f(x)=x^2, target 9 -> x*=3."""
        x, ind = binary_search(0.0, 10.0, 9.0, lambda x: x * x)
        assert ind == 0
        assert abs(x - 3.0) <= 0.01
        assert abs(x * x - 9.0) <= 0.1


# ---------------------------------------------------------------------------
# Core-layer vectors (pkg/core/accelerator_test.go, pkg/core/allocation_test.go)
# ---------------------------------------------------------------------------

from types import SimpleNamespace

from inferno_amd.config import (
    AcceleratorSpec,
    DecodeParms as CfgDecodeParms,
    ModelAcceleratorPerfData,
    PowerSpec,
    PrefillParms as CfgPrefillParms,
)
from inferno_amd.core.allocation import Allocation, _zero_load_allocation
from inferno_amd.core.system import Accelerator, Model


class TestAcceleratorPower:
    """Ref: accelerator_test.go TestAccelerator_Power — piecewise-linear."""

    @pytest.mark.parametrize(
        "util,want",
        [
            (0.0, 100.0),   # idle power
            (0.5, 300.0),   # mid power
            (1.0, 700.0),   # full power
            (0.25, 200.0),  # interpolated idle..mid
            (0.75, 500.0),  # interpolated mid..full
        ],
    )
    def test_vectors(self, util, want):
        acc = Accelerator(AcceleratorSpec(
            name="TestAcc",
            power=PowerSpec(idle=100, midPower=300, full=700, midUtil=0.5),
        ))
        acc.calculate()
        assert acc.power(util) == pytest.approx(want)


class TestTransitionPenalty:
    """Ref: allocation_test.go TestAllocation_TransitionPenalty."""

    def test_same_accelerator_same_replicas(self):
        a = Allocation(accelerator="gpu-a", num_replicas=2, cost=100.0)
        b = Allocation(accelerator="gpu-a", num_replicas=2, cost=100.0)
        assert a.transition_penalty(b) == 0.0

    def test_same_accelerator_different_replicas(self):
        a = Allocation(accelerator="gpu-a", num_replicas=2, cost=100.0)
        b = Allocation(accelerator="gpu-a", num_replicas=3, cost=150.0)
        assert a.transition_penalty(b) == pytest.approx(50.0)  # cost difference

    def test_different_accelerator(self):
        a = Allocation(accelerator="gpu-a", num_replicas=2, cost=100.0)
        b = Allocation(accelerator="gpu-b", num_replicas=2, cost=120.0)
        # 0.1*(100+120) + (120-100)
        assert a.transition_penalty(b) == pytest.approx(0.1 * 220.0 + 20.0)


class TestZeroLoadAllocation:
    """Ref: allocation_test.go TestZeroLoadAllocation."""

    @staticmethod
    def _model(name, instances):
        m = Model(name)
        m.num_instances = dict(instances)
        return m

    def test_zero_replicas_is_empty(self):
        server = SimpleNamespace(min_num_replicas=0, max_batch_size=0)
        model = self._model("test-model", {})
        acc = Accelerator(AcceleratorSpec(name="test-gpu", cost=100.0))
        perf = ModelAcceleratorPerfData(
            maxBatchSize=16,
            decodeParms=CfgDecodeParms(alpha=5.0, beta=2.0),
            prefillParms=CfgPrefillParms(gamma=10.0, delta=1.5),
        )
        alloc = _zero_load_allocation(server, model, acc, perf)
        assert alloc.accelerator == ""
        assert alloc.num_replicas == 0
        assert alloc.batch_size == 0
        assert alloc.cost == 0.0
        assert alloc.value == alloc.cost
        assert alloc.rho == 0

    def test_min_replicas(self):
        server = SimpleNamespace(min_num_replicas=2, max_batch_size=0)
        model = self._model("test-model", {"test-gpu": 1})
        acc = Accelerator(AcceleratorSpec(name="test-gpu", cost=100.0))
        perf = ModelAcceleratorPerfData(
            maxBatchSize=16,
            decodeParms=CfgDecodeParms(alpha=5.0, beta=2.0),
            prefillParms=CfgPrefillParms(gamma=10.0, delta=1.5),
        )
        alloc = _zero_load_allocation(server, model, acc, perf)
        assert alloc.accelerator == "test-gpu"
        assert alloc.num_replicas == 2
        assert alloc.batch_size == 16
        assert alloc.cost == pytest.approx(200.0)  # 100 * 1 instance * 2 replicas
        assert alloc.value == alloc.cost
        assert alloc.rho == 0
        assert alloc.itl == pytest.approx(5.0 + 2.0)      # alpha + beta
        assert alloc.ttft == pytest.approx(10.0 + 1.5)    # gamma + delta
        max_decode = 5.0 + 2.0 * 16
        max_serv = (10.0 + 1.5) + max_decode
        assert alloc.max_arrv_rate_per_replica == pytest.approx(16.0 / max_serv)

    def test_server_batch_override(self):
        server = SimpleNamespace(min_num_replicas=1, max_batch_size=8)
        model = self._model("test-model", {"test-gpu": 2})
        acc = Accelerator(AcceleratorSpec(name="test-gpu", cost=50.0))
        perf = ModelAcceleratorPerfData(
            maxBatchSize=16,  # overridden by server.max_batch_size
            decodeParms=CfgDecodeParms(alpha=3.0, beta=1.0),
            prefillParms=CfgPrefillParms(gamma=8.0, delta=2.0),
        )
        alloc = _zero_load_allocation(server, model, acc, perf)
        assert alloc.batch_size == 8
        assert alloc.num_replicas == 1
        assert alloc.cost == pytest.approx(100.0)  # 50 * 2 instances * 1 replica

    def test_minimal_inputs_do_not_crash(self):
        server = SimpleNamespace(min_num_replicas=1, max_batch_size=0)
        model = self._model("m", {})
        acc = Accelerator(AcceleratorSpec(name="test-gpu", cost=0.0))
        perf = ModelAcceleratorPerfData(
            maxBatchSize=1,
            decodeParms=CfgDecodeParms(alpha=0.1, beta=0.1),
            prefillParms=CfgPrefillParms(gamma=0.1, delta=0.1),
        )
        assert _zero_load_allocation(server, model, acc, perf) is not None


class TestServerPriority:
    """Ref: server_test.go TestServer_Priority + serviceclass clamping."""

    def _system(self):
        from inferno_amd.core.system import ServiceClass, System

        system = System()
        system.service_classes["high-priority"] = ServiceClass("high-priority", 1)
        system.service_classes["low-priority"] = ServiceClass("low-priority", 8)
        return system

    @pytest.mark.parametrize(
        "klass,expected",
        [
            ("high-priority", 1),
            ("low-priority", 8),
            ("nonexistent", 100),  # DefaultServiceClassPriority = DefaultLowPriority
        ],
    )
    def test_priority(self, klass, expected):
        from inferno_amd.config import AllocationData, ServerSpec
        from inferno_amd.core.system import Server

        server = Server(ServerSpec(
            name="test-server", model="test-model", klass=klass,
            currentAlloc=AllocationData(),
        ))
        assert server.priority(self._system()) == expected

    def test_out_of_range_priority_clamps_to_default(self):
        from inferno_amd.core.system import ServiceClass

        # priority outside [DefaultHighPriority=1, DefaultLowPriority=100]
        assert ServiceClass("weird", 0).priority == 100
        assert ServiceClass("weird", 101).priority == 100
        assert ServiceClass("ok", 50).priority == 50
