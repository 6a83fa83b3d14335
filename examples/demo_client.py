#!/usr/bin/env python3
"""End-to-end demo against a running gateway.

Start the server first (CPU works for a smoke; MI355X for real serving):

    python -m llm_instance_gateway_amd.server --model tiny --devices cpu &
    python examples/demo_client.py --base http://127.0.0.1:8081

Walks the whole API surface: completion (greedy + seeded sampling +
logprobs + stop), streaming, adapter hot-load and adapter-targeted
completion, model registration with a traffic split, and metrics.
(`tests/test_demo_example.py` runs this same walk hermetically.)
"""

import argparse
import json

from llm_instance_gateway_amd.api.client import GatewayClient


def run(gw: GatewayClient, model: str = "") -> None:
    models = gw.models()
    print("models:", [m["id"] for m in models])
    model = model or models[0]["id"]

    out = gw.completion(model, "hello world", max_tokens=12)
    print("greedy:", json.dumps(out["choices"][0]["text"])[:60], out["usage"])

    out = gw.completion(model, "hello world", max_tokens=12,
                        temperature=0.8, seed=7, logprobs=2)
    lp = out["choices"][0]["logprobs"]["token_logprobs"]
    print("seeded sampling, first logprobs:", [round(x, 3) for x in lp[:4]])

    out = gw.completion(model, "stop demo", max_tokens=32, stop=["\n\n"])
    print("finish_reason:", out["choices"][0]["finish_reason"])

    n_chunks = 0
    for chunk in gw.completion_stream(model, "streaming", max_tokens=8):
        c = chunk["choices"][0]
        n_chunks += 1
        if c.get("finish_reason"):
            print(f"stream: {n_chunks} chunks | finish:", c["finish_reason"])

    # adapter lifecycle (vLLM-compatible API; random-init weights offline)
    gw.load_adapter("demo-adapter")
    out = gw.completion(model, "with adapter", max_tokens=8, lora="demo-adapter")
    print("adapter completion tokens:", len(out["choices"][0]["token_ids"]))
    gw.unload_adapter("demo-adapter")

    # register a client model with a weighted split to the adapter
    gw.register_model("demo-split", criticality="Sheddable",
                      target_models=[{"name": "demo-adapter", "weight": 100}])
    print("registered:", [m["model_name"] for m in gw.list_inference_models()])
    gw.delete_model("demo-split")

    for line in gw.metrics_text().splitlines():
        if line.startswith("gateway_requests_total"):
            print("metric:", line)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--base", default="http://127.0.0.1:8081")
    ap.add_argument("--model", default="", help="client model name (default: first registered)")
    args = ap.parse_args()
    run(GatewayClient(args.base), args.model)


if __name__ == "__main__":
    main()
