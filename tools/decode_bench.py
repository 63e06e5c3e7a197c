"""Serving decode latency: eager KV-cached loop vs hipGraph-captured decoder.

Transformer-base, greedy decode.  The eager loop is launch-bound at small
batch (hundreds of kernel launches per token); GraphedDecoder replays one
captured graph per token.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from transformer_amd.models import Transformer
from transformer_amd.models.transformer import GraphedDecoder, greedy_decode

torch.manual_seed(0)
m = Transformer(num_layers=6, d_model=512, num_heads=8, dff=2048,
                input_vocab_size=32770, target_vocab_size=32770,
                rate=0.0, max_position=4096).cuda().bfloat16()
m.eval()

for B, S, L in [(1, 64, 256), (8, 64, 256), (16, 128, 128), (64, 256, 64)]:
    inp = torch.randint(2, 32768, (B, S), device="cuda")
    inp[:, 0] = 32768

    # eager
    greedy_decode(m, inp, 32768, 32769, max_len=8)  # warm
    torch.cuda.synchronize(); t0 = time.perf_counter()
    out_e = greedy_decode(m, inp, 32768, 32769, max_len=L)
    torch.cuda.synchronize(); te = time.perf_counter() - t0
    n_e = out_e.shape[1] - 1

    # graphed
    dec = GraphedDecoder(m, B=B, S_src=S, max_len=L, start_id=32768,
                         device=torch.device("cuda"))
    dec(inp, 32769, max_len=8)  # warm
    torch.cuda.synchronize(); t0 = time.perf_counter()
    out_g = dec(inp, 32769, max_len=L)
    torch.cuda.synchronize(); tg = time.perf_counter() - t0

    # parity up to first EOS per row
    ok = True
    for b in range(B):
        er, gr = out_e[b].tolist(), out_g[b].tolist()
        for i in range(min(len(er), len(gr))):
            if er[i] != gr[i]:
                ok = False
                break
            if er[i] == 32769:
                break
    print(f"B={B} S={S} L={L}: eager {te*1e3/n_e:.2f} ms/tok "
          f"({B*n_e/te:.0f} tok/s) | graphed {tg*1e3/L:.3f} ms/tok "
          f"({B*L/tg:.0f} tok/s) | speedup {te/n_e/(tg/L):.2f}x "
          f"| parity {'OK' if ok else 'MISMATCH'}")
print("decode bench done")
