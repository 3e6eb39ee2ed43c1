"""Attribute the at::native residue kernels (reduce_kernel / vectorized_elementwise)."""
import sys, torch
sys.path.insert(0, "/root/repo")
import bench as bench_mod
bench_mod._enable_tuned_gemms()
from libai_amd.models import GPTForPreTraining
from libai_amd.optim import FusedAdamW, get_default_optimizer_params
from libai_amd.utils import distributed as du
du.setup_dist_util({})
torch.manual_seed(0)
m = GPTForPreTraining(hidden_layers=24, vocab_size=50304, hidden_size=1024,
                      ffn_hidden_size=4096, num_attention_heads=16,
                      max_seq_length=1024, embedding_dropout_prob=0.1,
                      attention_dropout_prob=0.1, output_dropout_prob=0.1
                      ).to(torch.bfloat16).cuda()
opt = FusedAdamW(get_default_optimizer_params(m, base_lr=3e-4), lr=3e-4,
                 weight_decay=0.01, clip_grad=1.0)
ids = torch.randint(0, 50304, (48, 1025), device="cuda")
def step():
    opt.zero_grad()
    out = m(input_ids=ids[:, :-1], labels=ids[:, 1:])
    out["lm_loss"].backward()
    opt.step()
for _ in range(3): step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    step()
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=28))
