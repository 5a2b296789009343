"""Continuous-batching serving example: queue requests, step the engine."""

import torch

from neuronx_distributed_inference_amd.serving import ServingEngine

# `model` = a loaded CausalLMBase (see generation_demo.py)


def serve(model, tokenizer, prompts):
    eng = ServingEngine(model, chunk_size=512,
                        eos_token_id=tokenizer.eos_token_id)
    ids = [eng.add_request(
        tokenizer(p, return_tensors="pt").input_ids[0], max_new_tokens=128)
        for p in prompts]
    results = eng.run()
    print(eng.metrics())  # TTFT/TPOT percentiles, prefix-hit rate
    return {rid: tokenizer.decode(results[rid]) for rid in ids}
