"""Unit tests for the HPA stand-in's metric parsing/actuation helpers
(inferno_amd/testing/hpa.py) — e2e covers the loop; these pin the parser."""
from inferno_amd.testing.hpa import read_desired


class TestReadDesired:
    def test_parses_gauges(self):
        text = """# HELP inferno_desired_replicas d
# TYPE inferno_desired_replicas gauge
inferno_desired_replicas{accelerator_type="A100",namespace="llm-d-sim",variant_name="vllme-deploy"} 7.0
inferno_desired_replicas{accelerator_type="MI355X",namespace="prod",variant_name="llama70b-deploy"} 12.0
inferno_current_replicas{accelerator_type="A100",namespace="llm-d-sim",variant_name="vllme-deploy"} 3.0
"""
        got = read_desired(text)
        assert got == {("llm-d-sim", "vllme-deploy"): 7,
                       ("prod", "llama70b-deploy"): 12}

    def test_ignores_malformed_and_other_series(self):
        text = """inferno_desired_replicas{namespace="ns"} 3.0
inferno_desired_replicas{variant_name="v1",namespace="ns"} 2.5
python_info{version="3.10"} 1.0
"""
        got = read_desired(text)
        # no variant_name -> skipped; 2.5 floors to int
        assert got == {("ns", "v1"): 2}

    def test_empty(self):
        assert read_desired("") == {}
