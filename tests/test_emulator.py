"""Emulator tests: discrete-event vLLM model (memory accounting, batching,
eviction, token timing), metric exposition wire-format, FastAPI endpoint."""
import pytest

from inferno_amd.emulator.metrics import make_registry
from inferno_amd.emulator.sim import Device, VLLMSim


class TestDevice:
    def test_capacity_and_allocate(self):
        d = Device(mem_size_mb=1000, kv_mb_per_token=4, usable_ratio=0.8)
        assert d.capacity_mb == 800
        assert d.fits(200)
        assert d.allocate(100)  # 400 MB
        assert d.used_mb == 400
        assert not d.allocate(200)  # would be 1200 MB
        d.free(50)
        assert d.used_mb == 200

    def test_utilization(self):
        d = Device(mem_size_mb=100, kv_mb_per_token=1, usable_ratio=1.0)
        d.allocate(25)
        assert d.utilization == pytest.approx(0.25)


class TestSim:
    def test_single_request_timing(self):
        sim = VLLMSim(decode_time_ms=50, prefill_time_ms=100)
        req = sim.submit(input_tokens=10, output_tokens=4)
        sim.run_until_idle()
        # first step includes prefill: TTFT = 0.1 + 0.05
        assert req.first_token_time == pytest.approx(0.15)
        # remaining 3 tokens at 50 ms each
        assert req.finish_time == pytest.approx(0.15 + 3 * 0.05)
        assert sim.success_total == 1
        assert sim.avg_ttft_s == pytest.approx(0.15)
        assert sim.avg_tpot_s == pytest.approx(0.05)
        assert sim.device.used_mb == 0.0  # all KV freed

    def test_continuous_batching_shares_steps(self):
        sim = VLLMSim(decode_time_ms=50, prefill_time_ms=100, max_batch_size=8)
        reqs = [sim.submit(5, 10) for _ in range(4)]
        sim.run_until_idle()
        # all ran concurrently: finish at prefill + 10 decodes
        for r in reqs:
            assert r.finish_time == pytest.approx(0.1 + 10 * 0.05)

    def test_max_batch_limits_admission(self):
        sim = VLLMSim(max_batch_size=2)
        for _ in range(5):
            sim.submit(1, 3)
        sim._admit()
        assert sim.num_requests_running == 2
        assert sim.num_requests_waiting == 3

    def test_memory_limits_admission(self):
        # each request needs ~101 tokens * 4MB; capacity 800MB fits one
        sim = VLLMSim(mem_size_mb=1000, kv_mb_per_token=4, max_batch_size=16)
        sim.submit(100, 2)
        sim.submit(100, 2)
        sim._admit()
        assert sim.num_requests_running == 1
        sim.run_until_idle()
        assert sim.success_total == 2

    def test_eviction_under_pressure(self):
        # two long generations outgrow memory -> one gets preempted
        sim = VLLMSim(mem_size_mb=400, kv_mb_per_token=4, usable_ratio=1.0,
                      max_batch_size=4)
        sim.submit(20, 40)
        sim.submit(20, 40)
        sim.run_until_idle()
        assert sim.success_total == 2
        assert sim.preemptions >= 1
        assert sim.device.used_mb == 0.0

    def test_fifo_order(self):
        sim = VLLMSim(max_batch_size=1)
        a = sim.submit(1, 2)
        b = sim.submit(1, 2)
        sim.run_until_idle()
        assert a.finish_time < b.finish_time


class TestMetricsExposition:
    def test_wire_names_and_values(self):
        from prometheus_client import generate_latest

        sim = VLLMSim(decode_time_ms=50, prefill_time_ms=100)
        sim.submit(10, 4)
        sim.run_until_idle()
        reg = make_registry(sim, "default/default", "ns1")
        text = generate_latest(reg).decode()
        assert 'vllm:request_success_total{model_name="default/default",namespace="ns1"} 1.0' in text
        assert "vllm:num_requests_running{" in text
        assert "vllm:request_prompt_tokens_sum{" in text
        assert "vllm:request_prompt_tokens_count{" in text
        assert "vllm:request_generation_tokens_sum{" in text
        assert "vllm:time_to_first_token_seconds_sum{" in text
        assert "vllm:time_per_output_token_seconds_count{" in text

    def test_collector_can_parse_ttft(self):
        sim = VLLMSim()
        sim.submit(10, 4)
        sim.run_until_idle()
        reg = make_registry(sim, "m", "")
        from prometheus_client import generate_latest

        text = generate_latest(reg).decode()
        # without namespace, only the model_name label appears
        assert 'vllm:time_to_first_token_seconds_sum{model_name="m"}' in text


class TestServerEndpoint:
    def test_chat_completions_roundtrip(self):
        import os

        os.environ["DECODE_TIME"] = "1"
        os.environ["PREFILL_TIME"] = "1"
        try:
            from fastapi.testclient import TestClient

            from inferno_amd.emulator.server import build_app

            app = build_app()
            with TestClient(app) as client:
                r = client.post(
                    "/v1/chat/completions",
                    json={
                        "model": "default/default",
                        "messages": [{"role": "user", "content": "hello world"}],
                        "max_tokens": 3,
                    },
                )
                assert r.status_code == 200
                body = r.json()
                assert body["usage"]["completion_tokens"] == 3
                m = client.get("/metrics")
                assert "vllm:request_success_total" in m.text
        finally:
            del os.environ["DECODE_TIME"]
            del os.environ["PREFILL_TIME"]

    def test_mi355x_profile_memory(self):
        import os

        os.environ["DEVICE_PROFILE"] = "MI355X"
        try:
            from inferno_amd.emulator.server import build_app

            app = build_app()
            assert app.state.sim.device.capacity_mb == pytest.approx(288000 * 0.8)
        finally:
            del os.environ["DEVICE_PROFILE"]

    def test_loadgen_schedule_parse(self):
        from inferno_amd.emulator.loadgen import parse_schedule

        assert parse_schedule("[[60, 30], [120, 90.5]]") == [(60.0, 30.0), (120.0, 90.5)]
