"""Whisper decode modes head-to-head: greedy (hipGraph-replayed) vs
beam 2/4 (per-beam KV caches) — throughput per decoded token on one
chunk. Capability evidence for LYRICS_ASR_BEAM_SIZE parity."""

import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.models.whisper import (WhisperConfig, WhisperModel,
                                          beam_decode, greedy_decode)


def main():
    model = WhisperModel(WhisperConfig()).to("cuda", torch.bfloat16).eval()
    mel = torch.randn(80, 3000, device="cuda", dtype=torch.bfloat16)
    # warmup (graph capture for greedy; algo search for beam GEMMs)
    greedy_decode(model, mel, max_tokens=16, repetition_penalty=1.0,
                  no_repeat_ngram=0)
    beam_decode(model, mel, beam=2, max_tokens=16)
    torch.cuda.synchronize()
    mel2 = torch.randn(80, 3000, device="cuda", dtype=torch.bfloat16)
    for label, fn in [
        ("greedy   ", lambda: greedy_decode(mel=mel2, model=model,
                                            max_tokens=128,
                                            repetition_penalty=1.0,
                                            no_repeat_ngram=0)),
        ("beam k=2 ", lambda: beam_decode(model, mel2, beam=2,
                                          max_tokens=128)),
        ("beam k=4 ", lambda: beam_decode(model, mel2, beam=4,
                                          max_tokens=128)),
    ]:
        t0 = time.perf_counter()
        toks = fn()
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        n = max(len(toks), 1)
        print(f"{label}: {n:4d} tokens in {dt*1000:6.0f} ms "
              f"({n/dt:6.1f} tok/s incl. encode)")


if __name__ == "__main__":
    main()
