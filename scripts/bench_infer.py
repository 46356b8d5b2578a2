"""Inference engine throughput smoke (decode tokens/s on llama-7b)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from colossalai_amd.inference import GenerationConfig, InferenceConfig, LLMEngine
from colossalai_amd.models import LLAMA_CONFIGS, LlamaForCausalLM

cfg = LLAMA_CONFIGS["llama-7b"]
with torch.device("meta"):
    model = LlamaForCausalLM(cfg)
model = model.to_empty(device="cuda").to(torch.bfloat16)
with torch.no_grad():
    for p in model.parameters():
        p.normal_(0, 0.02)

engine = LLMEngine(model, InferenceConfig(max_batch_size=8, max_input_len=512, max_output_len=256))
prompts = [[int(x) for x in torch.randint(0, 32000, (128,))] for _ in range(8)]
# warmup
engine.generate(prompts, GenerationConfig(max_new_tokens=8))
torch.cuda.synchronize()
t0 = time.perf_counter()
N = 128
out = engine.generate(prompts, GenerationConfig(max_new_tokens=N))
torch.cuda.synchronize()
dt = time.perf_counter() - t0
total_new = sum(len(o) - 128 for o in out)
print(f"llama-7b bf16 bs8: {total_new} tokens in {dt:.2f}s = {total_new/dt:.1f} tok/s ({total_new/dt/8:.1f}/seq)")

# ---- paged continuous-batching engine, same workload + ragged arrivals
from colossalai_amd.inference import ContinuousBatchEngine

cengine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=8, max_input_len=512,
                                                       max_output_len=256))
cengine.generate(prompts, GenerationConfig(max_new_tokens=8))  # warmup
torch.cuda.synchronize()
t0 = time.perf_counter()
out = cengine.generate(prompts, GenerationConfig(max_new_tokens=N))
torch.cuda.synchronize()
dt = time.perf_counter() - t0
total_new = sum(len(o) - 128 for o in out)
print(f"llama-7b bf16 bs8 paged: {total_new} tokens in {dt:.2f}s = {total_new/dt:.1f} tok/s")

# ragged: 16 requests with varying lengths through 8 slots (continuous batching)
lens = [32, 200, 64, 120, 48, 256, 16, 96] * 2
rag_prompts = [[int(x) for x in torch.randint(0, 32000, (64,))] for _ in lens]
torch.cuda.synchronize()
t0 = time.perf_counter()
for p, n in zip(rag_prompts, lens):
    cengine.add_request(p, n)
cengine._gen = GenerationConfig(max_new_tokens=max(lens))
done = {}
while cengine.rm.has_work:
    done.update(cengine.step())
torch.cuda.synchronize()
dt = time.perf_counter() - t0
total_new = sum(len(t) - 64 for t in done.values())
print(f"llama-7b bf16 16-req ragged (8 slots): {total_new} tokens in {dt:.2f}s = {total_new/dt:.1f} tok/s")

# ---- hipGraph-captured decode loop
gengine = LLMEngine(model, InferenceConfig(max_batch_size=8, max_input_len=512, max_output_len=256,
                                           use_hip_graph=True))
gengine.generate(prompts, GenerationConfig(max_new_tokens=8))  # warmup + capture
torch.cuda.synchronize()
t0 = time.perf_counter()
out = gengine.generate(prompts, GenerationConfig(max_new_tokens=N))
torch.cuda.synchronize()
dt = time.perf_counter() - t0
total_new = sum(len(o) - 128 for o in out)
print(f"llama-7b bf16 bs8 hipGraph: {total_new} tokens in {dt:.2f}s = {total_new/dt:.1f} tok/s")


# ----- GQA model (llama3-8b: Hq32/Hkv8) — exercises the grouped decode kernel
cfg3 = LLAMA_CONFIGS["llama3-8b"]
with torch.device("meta"):
    model3 = LlamaForCausalLM(cfg3)
model3 = model3.to_empty(device="cuda").to(torch.bfloat16)
with torch.no_grad():
    for p_ in model3.parameters():
        p_.normal_(0, 0.02)
eng3 = LLMEngine(model3, InferenceConfig(max_batch_size=8, max_input_len=256, max_output_len=160))
out = eng3.generate(prompts, GenerationConfig(max_new_tokens=8))  # warm
t0 = time.perf_counter()
out = eng3.generate(prompts, GenerationConfig(max_new_tokens=128))
torch.cuda.synchronize()
dt = time.perf_counter() - t0
total_new = sum(len(o) - len(p) for o, p in zip(out, prompts))
print(f"llama3-8b (GQA 32/8) bf16 bs8: {total_new} tokens in {dt:.2f}s = {total_new/dt:.1f} tok/s")
