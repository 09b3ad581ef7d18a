"""inferno_amd — MI355X-native workload-variant autoscaler.

A ground-up rebuild of the capabilities of
llm-d-incubation/workload-variant-autoscaler ("Inferno" / WVA):

  * wire-compatible ``llmd.ai/v1alpha1 VariantAutoscaling`` CRD, Prometheus
    metric schema (``inferno_*`` gauges / ``vllm:*`` inputs) and HPA/KEDA
    actuation path;
  * the queueing-theory performance evaluator (state-dependent M/M/1/K chain,
    reference ``pkg/analyzer/``) and the global cost/SLO allocation solver
    (reference ``pkg/solver/``) re-designed as batched, log-space math that
    runs either vectorized on CPU or as hand-written HIP/CDNA4 kernels that
    sweep the full (model x accelerator x TP-degree) space per reconcile on
    one or more AMD Instinct MI355X GPUs;
  * an MI355X-first performance model (MFMA roofline, 288 GB HBM3E KV-cache
    sizing, RCCL-over-xGMI TP scaling) replacing NVIDIA SKU profile tables.

Layer map (mirrors reference SURVEY.md section 1):
  api/         CRD types + conditions            (ref: api/v1alpha1/)
  config/      JSON spec tree + defaults         (ref: pkg/config/)
  core/        system/server/allocation domain   (ref: pkg/core/)
  analyzer/    queueing models + search          (ref: pkg/analyzer/)
  solver/      unlimited + greedy solvers        (ref: pkg/solver/, pkg/manager/)
  engine/      batched SoA sweep engine (new)    (replaces scalar per-server loop)
  ops/         HIP/CDNA4 kernels + bindings (new)
  parallel/    multi-GPU sharding over RCCL (new)
  controller/  reconciler/collector/actuator     (ref: internal/*)
  perfmodel/   MI355X profile derivation (new)
  emulator/    vLLM discrete-event test double   (ref: tools/vllm-emulator/)
"""

__version__ = "0.1.0"
