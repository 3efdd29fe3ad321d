"""Sampling perf probe: 256 rows x 128k vocab (roadmap item 7 shape).
A/B the chunk-sum hierarchy sampler against torch and time each mode."""
import time

import torch

import flashinfer_amd as fi


def timeit(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


B, V = 256, 128256
torch.manual_seed(0)
logits = torch.randn(B, V, device="cuda") * 4.0
probs = torch.softmax(logits, dim=-1)

print(f"top_k(50)  : {timeit(lambda: fi.top_k_sampling_from_probs(probs, 50)):7.3f} ms")
print(f"top_p(0.9) : {timeit(lambda: fi.top_p_sampling_from_probs(probs, 0.9)):7.3f} ms")
print(f"topk_topp  : {timeit(lambda: fi.top_k_top_p_sampling_from_probs(probs, 50, 0.9)):7.3f} ms")
print(f"min_p(0.05): {timeit(lambda: fi.min_p_sampling_from_probs(probs, 0.05)):7.3f} ms")
print(f"plain      : {timeit(lambda: fi.sampling_from_probs(probs)):7.3f} ms")
print(f"logits topk50: {timeit(lambda: fi.top_k_top_p_sampling_from_logits(logits, 50, 1.0)):7.3f} ms")
print(f"torch.multinomial: {timeit(lambda: torch.multinomial(probs, 1)):7.3f} ms")

from flashinfer_amd import sampling
print(f"top_k_renorm(50) : {timeit(lambda: sampling.top_k_renorm_probs(probs, 50)):7.3f} ms")
print(f"top_p_renorm(0.9): {timeit(lambda: sampling.top_p_renorm_probs(probs, 0.9)):7.3f} ms")
print(f"top_k_mask_logits: {timeit(lambda: sampling.top_k_mask_logits(logits, 50)):7.3f} ms")
