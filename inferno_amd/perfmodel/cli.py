"""CLI: derive MI355X perf profiles for a model.

  python -m inferno_amd.perfmodel.cli llama-3.1-70b MI355X [--tps 1,2,4,8]
                                                           [--at-tokens 1024]
Prints the CR modelProfile.accelerators block as YAML.
"""
from __future__ import annotations

import argparse

import yaml

from . import (AMD_GPUS, DEEPSEEK_V3, GRANITE_13B, LLAMA_8B, LLAMA_70B, LLAMA_405B,
               LLAMA_405B_FP8, MIXTRAL_8X7B, QWEN_72B, LlmSpec, derive_profile,
               tp_variant_name)

KNOWN_MODELS = {
    m.name: m
    for m in (LLAMA_8B, LLAMA_70B, LLAMA_405B, LLAMA_405B_FP8, GRANITE_13B,
              MIXTRAL_8X7B, QWEN_72B, DEEPSEEK_V3)
}


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("model", help=f"one of {sorted(KNOWN_MODELS)} or params_b:layers:hidden:heads:kv_heads")
    p.add_argument("gpu", choices=sorted(AMD_GPUS), default="MI355X")
    p.add_argument("--tps", default="1,2,4,8")
    p.add_argument("--at-tokens", type=int, default=1024)
    args = p.parse_args()

    if args.model in KNOWN_MODELS:
        model = KNOWN_MODELS[args.model]
    else:
        parts = args.model.split(":")
        if len(parts) != 5:
            raise SystemExit(f"unknown model {args.model!r}")
        model = LlmSpec(
            name=f"custom-{parts[0]}b",
            params_b=float(parts[0]),
            layers=int(parts[1]),
            hidden=int(parts[2]),
            heads=int(parts[3]),
            kv_heads=int(parts[4]),
        )
    gpu = AMD_GPUS[args.gpu]
    out = []
    for tp in (int(t) for t in args.tps.split(",")):
        prof = derive_profile(model, gpu, tp, at_tokens=args.at_tokens)
        if prof is None:
            continue
        parms = prof.perf_parms()
        out.append(
            {
                "acc": tp_variant_name(gpu, tp),
                "accCount": tp,
                "perfParms": {
                    "decodeParms": parms.decodeParms,
                    "prefillParms": parms.prefillParms,
                },
                "maxBatchSize": prof.max_batch_size,
            }
        )
    print(yaml.safe_dump({"accelerators": out}, sort_keys=False))


if __name__ == "__main__":
    main()
