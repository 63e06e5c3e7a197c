import sys, torch
sys.path.insert(0, "/root/repo")
from transformer_amd.models import Transformer
from transformer_amd.models.transformer import greedy_decode

torch.manual_seed(0)
m = Transformer(num_layers=6, d_model=512, num_heads=8, dff=2048,
                input_vocab_size=32770, target_vocab_size=32770,
                rate=0.0, max_position=4096).cuda().bfloat16()
import time
for B, S, L in [(1, 64, 256), (16, 128, 128), (64, 256, 64)]:
    inp = torch.randint(2, 32768, (B, S), device="cuda")
    inp[:, 0] = 32768
    torch.cuda.synchronize(); t0 = time.perf_counter()
    out = greedy_decode(m, inp, 32768, 32769, max_len=L)
    torch.cuda.synchronize(); dt = time.perf_counter() - t0
    n_tok = out.shape[1] - 1
    print(f"B={B} S={S} decoded {n_tok} steps in {dt*1e3:.1f} ms "
          f"({B*n_tok/dt:.0f} tok/s, {dt*1e3/n_tok:.2f} ms/step)")
print("decode stress OK")
