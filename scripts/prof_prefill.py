import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_inference_amd.config import EngineConfig, OnDeviceSamplingConfig
from neuronx_distributed_inference_amd.models.llama.modeling_llama import (
    LlamaForCausalLM, LlamaInferenceConfig)

B, CTX = 128, 128
ec = EngineConfig(batch_size=B, max_batch_size=B, seq_len=CTX + 8, max_length=CTX + 8,
                  torch_dtype="bfloat16", tp_degree=1,
                  on_device_sampling_config=OnDeviceSamplingConfig(top_k=1),
                  enable_hip_graphs=False, fused_qkv=True, fused_gate_up=True)
cfg = LlamaInferenceConfig(neuron_config=ec, max_position_embeddings=4096,
    rms_norm_eps=1e-5, rope_theta=500000.0, hidden_act="silu",
    tie_word_embeddings=False, hidden_size=4096, intermediate_size=14336,
    num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
    vocab_size=128256)
with torch.device("cuda"):
    m = LlamaForCausalLM(cfg).eval()
with torch.no_grad():
    for p in m.parameters():
        if p.dim() >= 2:
            p.normal_(0, 0.02)
ids = torch.randint(0, 128256, (B, CTX), device="cuda")
pos = torch.arange(CTX, device="cuda").view(1, -1).expand(B, -1).contiguous()
sq = torch.arange(B, device="cuda")
sp = None  # model memoizes default sampling params internally
with torch.inference_mode():
    for _ in range(2):
        m.reset(); m(ids, None, pos, sq, sp)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        m.reset(); m(ids, None, pos, sq, sp)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 5
tok = B * CTX
print(f"prefill {B}x{CTX}: {dt*1e3:.1f} ms, {tok/dt:.0f} tok/s, "
      f"{2*8.03e9*tok/dt/1e15:.2f} PF/s")
