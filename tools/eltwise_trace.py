"""Identify which aten ops spend GPU time in the train step (GPU box)."""
import sys
import torch

sys.path.insert(0, ".")
from transformer_amd.models import Transformer
from transformer_amd.runtime import NoamAdam
from transformer_amd import ops

torch.manual_seed(0)
model = Transformer(num_layers=6, d_model=512, num_heads=8, dff=2048,
                    input_vocab_size=32770, target_vocab_size=32770,
                    rate=0.1, max_position=4096).cuda().bfloat16()
opt = NoamAdam(model, 512, use_flat=True)
B, S = 64, 256
src = torch.randint(2, 32768, (B, S), device="cuda")
tar = torch.randint(2, 32768, (B, S), device="cuda")


def step():
    ti, tr = tar[:, :-1].contiguous(), tar[:, 1:].contiguous()
    logits, _ = model((src, ti), training=True)
    loss = ops.masked_cross_entropy(logits, tr, B, 0.1)
    opt.zero_grad()
    loss.backward()
    opt.step()


for _ in range(3):
    step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as p:
    for _ in range(3):
        step()
    torch.cuda.synchronize()
print(p.key_averages().table(sort_by="self_cuda_time_total", row_limit=28,
                             max_name_column_width=55))
