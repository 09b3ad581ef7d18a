"""MI355X-first performance-model derivation.

Replaces the reference's fitted NVIDIA-SKU profile tables
(docs/design/modeling-optimization.md:52-65: L40S/L4/H100/A100 + tp 1/2/4/8
benchmark grids) with an analytical CDNA4 model that derives the queueing
evaluator's (alpha, beta, gamma, delta, maxBatchSize, atTokens) per
(LLM, accelerator, TP-degree) from first principles:

  * decode ITL(b) = alpha + beta*b — decode is HBM-bound on CDNA4: alpha is
    the weight-streaming floor (param bytes / TP over achievable HBM3E
    bandwidth, ~6.3 of 8 TB/s peak) plus the per-layer RCCL all-reduce floor
    when TP>1; beta adds the per-request KV-read and MFMA compute slopes.
  * prefill TTFT(L, b) = gamma + delta*L*b — prefill is MFMA-bound:
    delta = 2*params/TP over the effective matrix throughput (bf16 dense
    ~2.5 PFLOP/s per GPU x an attainable-efficiency factor), gamma is the
    scheduling + collective launch floor.
  * TP scaling uses a ring-all-reduce cost over POINT-TO-POINT xGMI links
    (7 links x ~153 GB/s per GPU — NOT an NVSwitch fat tree): per all-reduce
    of message M over G GPUs, t = 2(G-1)/G * M / link_bw + (G-1)*hop_lat.
  * maxBatchSize from 288 GB HBM3E KV-cache sizing: usable = cap*0.9 -
    param_bytes/TP; kv/token = 2*layers*kv_heads*head_dim*bytes/TP;
    maxBatchSize = usable / (kv_per_token * atTokens).

Profiles are emitted in the exact CR ``PerfParms`` string format
(api/v1alpha1: decodeParms {"alpha","beta"}, prefillParms {"gamma","delta"})
and as ``ModelAcceleratorPerfData`` rows with TP-degree as a first-class
variant axis (accelerator name "<ACC>-TP<G>").
"""
from __future__ import annotations

from dataclasses import dataclass, field

from ..api import v1alpha1 as api
from ..config import AcceleratorSpec, DecodeParms, ModelAcceleratorPerfData, PowerSpec, PrefillParms


@dataclass
class GpuSpec:
    """Accelerator hardware description (per GPU)."""

    name: str
    mem_gb: float
    hbm_peak_tbs: float  # peak HBM bandwidth, TB/s
    hbm_eff: float  # achievable fraction
    bf16_pflops: float  # dense MFMA peak, PFLOP/s
    fp8_pflops: float
    mfma_eff: float  # attainable fraction in prefill GEMMs
    link_gbs: float  # per-link xGMI bandwidth, GB/s
    n_links: int
    hop_latency_us: float
    cost_cents_hr: float
    power: PowerSpec = field(default_factory=PowerSpec)


# CDNA4 / MI355X numbers (MI355X_MICROARCH.md; achievable fractions are
# conservative engineering estimates, not AMD's 2:1-sparse marketing peaks)
MI355X = GpuSpec(
    name="MI355X",
    mem_gb=288.0,
    hbm_peak_tbs=8.0,
    hbm_eff=0.79,  # ~6.3 TB/s achievable
    bf16_pflops=2.5,
    fp8_pflops=5.0,
    mfma_eff=0.45,
    link_gbs=153.0,
    n_links=7,
    hop_latency_us=2.0,
    cost_cents_hr=95.0,
    power=PowerSpec(idle=180, full=1400, midPower=950, midUtil=0.55),
)

MI325X = GpuSpec(
    name="MI325X",
    mem_gb=256.0,
    hbm_peak_tbs=6.0,
    hbm_eff=0.75,
    bf16_pflops=1.3,
    fp8_pflops=2.6,
    mfma_eff=0.42,
    link_gbs=128.0,
    n_links=7,
    hop_latency_us=2.0,
    cost_cents_hr=78.0,
    power=PowerSpec(idle=150, full=1000, midPower=700, midUtil=0.6),
)

MI300X = GpuSpec(
    name="MI300X",
    mem_gb=192.0,
    hbm_peak_tbs=5.3,
    hbm_eff=0.72,
    bf16_pflops=1.3,
    fp8_pflops=2.6,
    mfma_eff=0.40,
    link_gbs=128.0,
    n_links=7,
    hop_latency_us=2.0,
    cost_cents_hr=65.0,
    power=PowerSpec(idle=140, full=750, midPower=550, midUtil=0.6),
)

AMD_GPUS = {"MI355X": MI355X, "MI325X": MI325X, "MI300X": MI300X}


@dataclass
class LlmSpec:
    """Transformer shape for the perf model.

    ``active_params_b`` covers MoE models: decode weight streaming reads only
    the routed experts' parameters per token (plus shared weights), while
    memory capacity must hold ``params_b`` (all experts) — so alpha/beta/delta
    derive from the ACTIVE parameter count and KV/memory sizing from the
    TOTAL. Dense models leave it None (= params_b). ``dtype_bytes`` 2.0 =
    bf16; 1.0 = OCP fp8 (gfx950's e4m3/e5m2 — weights AND the MFMA rate
    double, hence the fp8_pflops column of GpuSpec).
    """

    name: str
    params_b: float  # TOTAL parameters, billions
    layers: int
    hidden: int
    heads: int
    kv_heads: int
    dtype_bytes: float = 2.0  # bf16 weights/KV; 1.0 = fp8
    active_params_b: float | None = None  # MoE: routed-active params

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads

    @property
    def active_b(self) -> float:
        return self.active_params_b if self.active_params_b is not None else self.params_b

    @property
    def param_bytes(self) -> float:
        """TOTAL weight bytes (memory capacity)."""
        return self.params_b * 1e9 * self.dtype_bytes

    @property
    def active_param_bytes(self) -> float:
        """ACTIVE weight bytes per token (decode streaming)."""
        return self.active_b * 1e9 * self.dtype_bytes

    def kv_bytes_per_token(self, tp: int) -> float:
        """KV cache bytes per token per TP shard: 2 (K and V) x layers x
        kv_heads x head_dim x dtype / tp."""
        return 2.0 * self.layers * self.kv_heads * self.head_dim * self.dtype_bytes / tp


# reference model shapes (public architecture parameters)
LLAMA_8B = LlmSpec("llama-3.1-8b", 8.0, 32, 4096, 32, 8)
LLAMA_70B = LlmSpec("llama-3.1-70b", 70.0, 80, 8192, 64, 8)
LLAMA_405B = LlmSpec("llama-3.1-405b", 405.0, 126, 16384, 128, 8)
LLAMA_405B_FP8 = LlmSpec("llama-3.1-405b-fp8", 405.0, 126, 16384, 128, 8,
                         dtype_bytes=1.0)
GRANITE_13B = LlmSpec("granite-13b", 13.0, 40, 5120, 40, 40)
MIXTRAL_8X7B = LlmSpec("mixtral-8x7b", 46.7, 32, 4096, 32, 8,
                       active_params_b=12.9)
QWEN_72B = LlmSpec("qwen2.5-72b", 72.7, 80, 8192, 64, 8)
DEEPSEEK_V3 = LlmSpec("deepseek-v3", 671.0, 61, 7168, 128, 128,
                      dtype_bytes=1.0, active_params_b=37.0)


def allreduce_ms(message_bytes: float, tp: int, gpu: GpuSpec) -> float:
    """Ring all-reduce cost over point-to-point xGMI (ms). Each step moves
    M/G per GPU over ONE link; 2(G-1) steps -> 2(G-1)/G * M / link_bw."""
    if tp <= 1:
        return 0.0
    bw = gpu.link_gbs * 1e9  # B/s
    t_bytes = 2.0 * (tp - 1) / tp * message_bytes / bw
    t_lat = (tp - 1) * gpu.hop_latency_us * 1e-6
    return (t_bytes + t_lat) * 1e3


@dataclass
class DerivedProfile:
    alpha: float  # ms
    beta: float  # ms per request
    gamma: float  # ms
    delta: float  # ms per (input token x batch)
    max_batch_size: int
    at_tokens: int
    acc_count: int  # = TP degree

    def perf_parms(self) -> api.PerfParms:
        """CR wire format (string maps)."""
        return api.PerfParms(
            decodeParms={"alpha": f"{self.alpha:.4f}", "beta": f"{self.beta:.6f}"},
            prefillParms={"gamma": f"{self.gamma:.4f}", "delta": f"{self.delta:.8f}"},
        )


def derive_profile(
    model: LlmSpec,
    gpu: GpuSpec,
    tp: int = 1,
    at_tokens: int = 1024,
    sched_overhead_ms: float = 2.0,
    kv_mem_fraction: float = 0.9,
) -> DerivedProfile | None:
    """Analytic (alpha,beta,gamma,delta,maxBatch) for (model, gpu, TP).

    Returns None when the model does not fit (param shard + minimal KV
    exceeds device memory).
    """
    hbm_bps = gpu.hbm_peak_tbs * 1e12 * gpu.hbm_eff
    pflops = gpu.fp8_pflops if model.dtype_bytes <= 1.0 else gpu.bf16_pflops
    mfma_flops = pflops * 1e15 * gpu.mfma_eff

    # ---- decode ITL = alpha + beta*b ---------------------------------
    # alpha: stream the ACTIVE weight shard once per token + per-layer
    # all-reduce (MoE models stream only routed experts)
    weight_ms = model.active_param_bytes / tp / hbm_bps * 1e3
    # decode all-reduce per layer: message = hidden * dtype (batch~1 row)
    ar_decode = model.layers * allreduce_ms(model.hidden * model.dtype_bytes, tp, gpu)
    alpha = weight_ms + ar_decode + sched_overhead_ms * 0.25
    # beta: per extra in-flight request, read its KV (at_tokens context) and
    # do its GEMV compute share
    kv_read_ms = model.kv_bytes_per_token(tp) * at_tokens / hbm_bps * 1e3
    compute_ms = 2.0 * model.active_b * 1e9 / tp / mfma_flops * 1e3
    beta = kv_read_ms + compute_ms

    # ---- prefill TTFT = gamma + delta * inTokens * b ------------------
    # compute-bound: 2*params FLOPs per token, MFMA at prefill efficiency.
    # Per-token all-reduce bandwidth is negligible vs the GEMM work, so the
    # collective cost appears only as the latency floor in gamma — keeping
    # the model linear exactly like the CR's gamma + delta*tokens*batch form.
    delta = 2.0 * model.active_b * 1e9 / tp / mfma_flops * 1e3
    gamma = sched_overhead_ms + model.layers * allreduce_ms(
        model.hidden * model.dtype_bytes, tp, gpu
    )

    # ---- KV-cache sizing -> maxBatchSize ------------------------------
    mem_bytes = gpu.mem_gb * 1e9
    usable = mem_bytes * kv_mem_fraction - model.param_bytes / tp
    if usable <= 0:
        return None
    kv_per_req = model.kv_bytes_per_token(tp) * at_tokens
    max_batch = int(usable // kv_per_req)
    if max_batch < 1:
        return None

    return DerivedProfile(
        alpha=alpha,
        beta=beta,
        gamma=gamma,
        delta=delta,
        max_batch_size=max_batch,
        at_tokens=at_tokens,
        acc_count=tp,
    )


def tp_variant_name(gpu: GpuSpec, tp: int) -> str:
    return gpu.name if tp == 1 else f"{gpu.name}-TP{tp}"


def accelerator_spec(gpu: GpuSpec, tp: int) -> AcceleratorSpec:
    """Accelerator table row for a TP variant (cost scales with GPUs)."""
    return AcceleratorSpec(
        name=tp_variant_name(gpu, tp),
        type=f"AMD-{gpu.name}-{int(gpu.mem_gb)}GB",
        multiplicity=1,
        memSize=int(gpu.mem_gb),
        memBW=int(gpu.hbm_peak_tbs * 1000),
        power=gpu.power,
        cost=gpu.cost_cents_hr,
    )


def derive_perf_data(
    model: LlmSpec,
    gpus: list[GpuSpec] | None = None,
    tps: tuple[int, ...] = (1, 2, 4, 8),
    at_tokens: int = 1024,
) -> list[ModelAcceleratorPerfData]:
    """Emit ModelAcceleratorPerfData rows: TP-degree is a first-class
    variant axis (one row per (gpu, tp) that fits)."""
    gpus = gpus if gpus is not None else [MI355X]
    rows: list[ModelAcceleratorPerfData] = []
    for gpu in gpus:
        for tp in tps:
            prof = derive_profile(model, gpu, tp, at_tokens=at_tokens)
            if prof is None:
                continue
            rows.append(
                ModelAcceleratorPerfData(
                    name=model.name,
                    acc=tp_variant_name(gpu, tp),
                    accCount=tp,
                    maxBatchSize=prof.max_batch_size,
                    atTokens=prof.at_tokens,
                    decodeParms=DecodeParms(alpha=prof.alpha, beta=prof.beta),
                    prefillParms=PrefillParms(gamma=prof.gamma, delta=prof.delta),
                )
            )
    return rows


def accelerator_profile_for_cr(model: LlmSpec, gpu: GpuSpec, tp: int,
                               at_tokens: int = 1024) -> api.AcceleratorProfile | None:
    """AcceleratorProfile block for a VariantAutoscaling CR spec."""
    prof = derive_profile(model, gpu, tp, at_tokens=at_tokens)
    if prof is None:
        return None
    return api.AcceleratorProfile(
        acc=tp_variant_name(gpu, tp),
        accCount=tp,
        perfParms=prof.perf_parms(),
        maxBatchSize=prof.max_batch_size,
    )
