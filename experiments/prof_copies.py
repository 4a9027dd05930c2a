import sys, os
sys.path.insert(0, "/root/repo")
os.environ.setdefault("LAH_FUSED_DW", "0")
import torch
from torch.profiler import profile, ProfilerActivity
sys.argv = ["bench.py", "--steps", "1", "--warmup", "1", "--no-dht"]
import runpy
# monkeypatch: run one profiled step via bench main is awkward; replicate inline
from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig
from learning_at_home_amd.parallel import ExpertParallelMoE, FusedExpertSGD
cfg = TransformerDMoEConfig(vocab_size=32000, d_model=1024, n_heads=16, n_layers=6,
                            seq_len=1024, num_experts=256, k_best=4, ffn_dim=4096)
dev = torch.device("cuda:0")
model = TransformerDMoE(cfg, lambda i,c: ExpertParallelMoE(
    d_model=c.d_model, ffn_dim=c.ffn_dim, num_experts=c.num_experts, k=c.k_best,
    capacity_factor=1.5, aux_loss_coef=0.01)).to(dev, dtype=torch.bfloat16)
for b in model.blocks:
    b.moe.experts.float(); b.moe.experts.init_gpu_state(master_weights=False)
from learning_at_home_amd.optim import MasterSGD
opts = [FusedExpertSGD(b.moe.experts, lr=0.01, stochastic_rounding=True) for b in model.blocks]
sstream = torch.cuda.Stream(device=dev)
for o in opts: o.set_side_stream(sstream)
trunk = [p for n, p in model.named_parameters() if ".moe.experts." not in n]
trunk_opt = MasterSGD(trunk, lr=0.01)
def step():
    ids = torch.randint(0, 32000, (96, 1025), device=dev)
    loss = model.loss(ids[:, :-1], ids[:, 1:], fused_ce=True)
    trunk_opt.zero_grad(set_to_none=True)
    loss.backward()
    for o in opts: o.step()
    trunk_opt.step()
step(); torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA], with_stack=True) as prof:
    step(); torch.cuda.synchronize()
evs = prof.key_averages(group_by_stack_n=6)
for ev in sorted(evs, key=lambda e: -e.self_device_time_total)[:60]:
    if "Memcpy" in ev.key or "copy" in ev.key.lower():
        print(f"{ev.self_device_time_total/1e3:.1f} ms  {ev.count}x  {ev.key}")
        if ev.stack:
            for fr in ev.stack[:6]:
                print("     ", fr)
