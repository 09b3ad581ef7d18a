"""Deploy-artifact tests: the shipped example VA + ConfigMaps drive a full
reconcile (the stage-1 oracle of SURVEY.md section 7: the reference's sample
VA validates and scales unchanged)."""
import json
import time

import pytest
import yaml

from inferno_amd.api import v1alpha1 as api
from inferno_amd.controller import adapters, collector
from inferno_amd.controller.collector import MockPromAPI, Sample
from inferno_amd.controller.k8s import Deployment, InMemoryKube
from inferno_amd.controller.metrics import MetricsEmitter
from inferno_amd.controller.reconciler import Reconciler
from prometheus_client import CollectorRegistry

NS = "workload-variant-autoscaler-system"


def load_yaml_docs(path):
    with open(path) as f:
        return [d for d in yaml.safe_load_all(f) if d]


class TestExampleArtifacts:
    def test_crd_schema_shape(self):
        crd = load_yaml_docs("deploy/crd/llmd.ai_variantautoscalings.yaml")[0]
        assert crd["metadata"]["name"] == "variantautoscalings.llmd.ai"
        spec = crd["spec"]
        assert spec["group"] == "llmd.ai"
        assert spec["names"]["shortNames"] == ["va"]
        v = spec["versions"][0]
        assert v["name"] == "v1alpha1" and v["served"] and v["storage"]
        props = v["schema"]["openAPIV3Schema"]["properties"]
        assert set(props["spec"]["required"]) == {"modelID", "sloClassRef", "modelProfile"}
        status = props["status"]["properties"]
        assert "currentAlloc" in status and "desiredOptimizedAlloc" in status
        assert "conditions" in status and "actuation" in status
        cols = {c["name"] for c in v["additionalPrinterColumns"]}
        assert cols == {"Model", "Accelerator", "CurrentReplicas", "Optimized",
                        "MetricsReady", "Age"}

    def test_example_va_parses(self):
        docs = load_yaml_docs("deploy/examples/vllme-variantautoscaling.yaml")
        for doc in docs:
            va = api.va_from_json(doc)
            assert va.spec.modelID
            assert va.accelerator_name
            for ap in va.spec.modelProfile.accelerators:
                float(ap.perfParms.decodeParms["alpha"])
                float(ap.perfParms.prefillParms["delta"])

    def test_example_configmaps_parse(self):
        acc_cm_doc = load_yaml_docs("deploy/configmap-accelerator-unitcost.yaml")[0]
        svc_cm_doc = load_yaml_docs("deploy/configmap-serviceclass.yaml")[0]
        spec = adapters.create_system_data(acc_cm_doc["data"], svc_cm_doc["data"])
        assert {a.name for a in spec.accelerators} >= {"A100", "MI300X", "MI355X"}
        assert {c.name for c in spec.serviceClasses} == {"Premium", "Freemium"}

    def test_reconcile_with_shipped_artifacts(self):
        """The shipped sample VA scales end-to-end against the shipped
        ConfigMaps (mock Prometheus supplying guidellm-like load)."""
        acc_cm = load_yaml_docs("deploy/configmap-accelerator-unitcost.yaml")[0]["data"]
        svc_cm = load_yaml_docs("deploy/configmap-serviceclass.yaml")[0]["data"]
        va_doc = load_yaml_docs("deploy/examples/vllme-variantautoscaling.yaml")[0]
        va = api.va_from_json(va_doc)

        kube = InMemoryKube()
        kube.add_configmap(NS, "accelerator-unit-costs", acc_cm)
        kube.add_configmap(NS, "service-classes-config", svc_cm)
        kube.add_configmap(NS, "workload-variant-autoscaler-variantautoscaling-config",
                           {"GLOBAL_OPT_INTERVAL": "60s"})
        kube.add_va(va)
        kube.add_deployment(
            Deployment(name=va.name, namespace=va.namespace, replicas=1,
                       status_replicas=1, uid="uid-ex")
        )
        now = time.time()
        model, ns = va.spec.modelID, va.namespace
        prom = MockPromAPI(
            results={
                collector.arrival_query(model, ns): [Sample(1.1, now)],
                collector.ttft_query(model, ns): [Sample(0.015, now)],
                collector.itl_query(model, ns): [Sample(0.007, now)],
                collector.avg_prompt_tokens_query(model, ns): [Sample(128, now)],
                collector.avg_decode_tokens_query(model, ns): [Sample(64, now)],
            }
        )
        rec = Reconciler(kube, prom, MetricsEmitter(registry=CollectorRegistry()),
                         backend="cpu", scale_to_zero=False)
        result = rec.reconcile()
        assert result.processed == 1
        stored = kube.vas[(va.namespace, va.name)]
        des = stored.status.desiredOptimizedAlloc
        assert des.accelerator == va.accelerator_name  # keepAccelerator
        assert des.numReplicas >= 1
        assert api.is_condition_true(stored, api.TYPE_OPTIMIZATION_READY)

    def test_helm_values_parse(self):
        vals = load_yaml_docs("charts/workload-variant-autoscaler/values.yaml")[0]
        assert vals["controller"]["backend"] in ("auto", "gpu", "cpu")
        chart = load_yaml_docs("charts/workload-variant-autoscaler/Chart.yaml")[0]
        assert chart["name"] == "workload-variant-autoscaler"


class TestOfflineCli:
    def test_solve_and_analyze(self, tmp_path, capsys):
        import json as _json
        import sys as _sys

        from inferno_amd.config import system_spec_to_json
        from inferno_amd import cli
        from tests.fixtures import make_spec

        path = tmp_path / "sys.json"
        path.write_text(_json.dumps(system_spec_to_json(make_spec(n_servers=3, seed=56))))

        class A:  # solve args
            spec = str(path)
            backend = "cpu"
            json = True

        assert cli.cmd_solve(A()) == 0
        out = _json.loads(capsys.readouterr().out)
        assert len(out["allocations"]) == 3

        class B:  # analyze args
            spec = str(path)
            backend = "cpu"
            server = "srv-1:ns"

        assert cli.cmd_analyze(B()) == 0
        assert "accelerator" in capsys.readouterr().out


class TestShippedSystemJson:
    def test_cli_solves_shipped_example(self, capsys):
        import json as _json

        from inferno_amd import cli

        class A:
            spec = "examples/system.json"
            backend = "cpu"
            json = True

        assert cli.cmd_solve(A()) == 0
        out = _json.loads(capsys.readouterr().out)
        assert len(out["allocations"]) == 6
        for d in out["allocations"].values():
            assert d["numReplicas"] >= 1
