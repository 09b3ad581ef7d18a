"""OpenAI-compatible FastAPI emulator server.

Ref tools/vllm-emulator/server.py:22-126: env-configured vLLM emulator with
POST /v1/chat/completions (waits for simulated completion) and /metrics in
Prometheus exposition format. Env: MODEL_NAME, NAMESPACE, DECODE_TIME (ms),
PREFILL_TIME (ms), MEM_SIZE (MB), MAX_BATCH_SIZE, KV_MB_PER_TOKEN,
AVG_OUTPUT_TOKENS; set DEVICE_PROFILE=MI355X for the 288 GB HBM3E profile.
"""

import asyncio
import os
import random
import time

from .metrics import make_registry
from .sim import MI355X_MEM_SIZE_MB, VLLMSim


def build_app():
    from fastapi import FastAPI, Request
    from fastapi.responses import PlainTextResponse
    from prometheus_client import generate_latest

    model_name = os.environ.get("MODEL_NAME", "default/default")
    namespace = os.environ.get("NAMESPACE", "")
    mem_size = float(os.environ.get("MEM_SIZE", "80000"))
    if os.environ.get("DEVICE_PROFILE", "").upper() == "MI355X":
        mem_size = MI355X_MEM_SIZE_MB
    sim = VLLMSim(
        decode_time_ms=float(os.environ.get("DECODE_TIME", "50")),
        prefill_time_ms=float(os.environ.get("PREFILL_TIME", "100")),
        mem_size_mb=mem_size,
        kv_mb_per_token=float(os.environ.get("KV_MB_PER_TOKEN", "4")),
        max_batch_size=int(os.environ.get("MAX_BATCH_SIZE", "256")),
    )
    avg_out = float(os.environ.get("AVG_OUTPUT_TOKENS", "64"))
    registry = make_registry(sim, model_name, namespace)

    async def run_loop():
        # advance the simulated scheduler in real time
        while True:
            if sim.waiting or sim.running:
                before = sim.clock
                sim.step()
                await asyncio.sleep(max(sim.clock - before, 0.0))
            else:
                await asyncio.sleep(sim.decode_time_s)

    from contextlib import asynccontextmanager

    @asynccontextmanager
    async def lifespan(app_):
        app_.state.loop_task = asyncio.create_task(run_loop())
        try:
            yield
        finally:
            app_.state.loop_task.cancel()

    app = FastAPI(title="inferno-amd vllm emulator", lifespan=lifespan)
    app.state.sim = sim

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        body = await request.json()
        messages = body.get("messages", [])
        prompt = " ".join(m.get("content", "") for m in messages)
        input_tokens = max(len(prompt.split()), 1)
        max_tokens = body.get("max_tokens")
        if max_tokens:
            output_tokens = int(max_tokens)
        else:
            output_tokens = max(int(random.expovariate(1.0 / avg_out)), 1)
        done = asyncio.Event()
        loop = asyncio.get_running_loop()
        req = sim.submit(
            input_tokens, output_tokens,
            on_finish=lambda r: loop.call_soon_threadsafe(done.set),
        )
        await done.wait()
        return {
            "id": f"cmpl-{req.rid}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": body.get("model", model_name),
            "choices": [
                {
                    "index": 0,
                    "message": {"role": "assistant", "content": "x " * req.generated},
                    "finish_reason": "stop",
                }
            ],
            "usage": {
                "prompt_tokens": req.input_tokens,
                "completion_tokens": req.generated,
                "total_tokens": req.input_tokens + req.generated,
            },
        }

    @app.get("/metrics")
    async def metrics():
        return PlainTextResponse(
            generate_latest(registry), media_type="text/plain; version=0.0.4"
        )

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app


def main():
    import uvicorn

    uvicorn.run(build_app(), host="0.0.0.0", port=int(os.environ.get("PORT", "8000")))


if __name__ == "__main__":
    main()
