"""External-bar check: time a HuggingFace
transformers LlamaForCausalLM training step (torch SDPA attention,
fused AdamW) on the SAME config/batch as the in-repo trainer, so the
perf-vs-real bar is not self-referential.

Megatron-LM is not installable in this image (no network); transformers
is, and its Llama with PyTorch SDPA (AOTriton flash) is an external,
independently-optimized implementation. Writes gpurun_out/external.json.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def hf_step_time(seq=4096, mbs=1, mbc=4, steps=3, warmup=2):
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig(
        hidden_size=4096, intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, vocab_size=128256,
        max_position_embeddings=seq, rope_theta=500000.0,
        attn_implementation="sdpa", use_cache=False)
    t0 = time.time()
    with torch.device("cuda"):
        model = LlamaForCausalLM(cfg).to(torch.bfloat16)
    model.gradient_checkpointing_disable()
    model.train()
    print(f"[hf] built {sum(p.numel() for p in model.parameters())/1e9:.2f}B "
          f"params in {time.time()-t0:.0f}s", flush=True)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4, betas=(0.9, 0.95),
                            eps=1e-8, foreach=False, fused=True)
    toks = torch.randint(0, cfg.vocab_size, (mbc, mbs, seq), device="cuda")

    def step():
        opt.zero_grad(set_to_none=True)
        for mb in range(mbc):
            out = model(input_ids=toks[mb], labels=toks[mb])
            out.loss.backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        opt.step()

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    torch.cuda.reset_peak_memory_stats()
    t0 = time.time()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    ms = (time.time() - t0) / steps * 1e3
    peak = torch.cuda.max_memory_allocated() / 2**30
    return ms, peak


def main():
    ms, peak = hf_step_time()
    tokens_s = 4 * 4096 / (ms / 1e3)
    out = dict(impl="transformers LlamaForCausalLM sdpa + fused AdamW",
               config="llama3-8b seq4096 mbs1 mbc4 bf16",
               ms_per_step=round(ms, 2), peak_gib=round(peak, 2),
               tokens_per_s=round(tokens_s, 1))
    print(json.dumps(out), flush=True)
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/external.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
