import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
import torch, numpy as np
from copy import deepcopy
from torch_actor_critic_amd.algo.graph import GraphedSACUpdate
from torch_actor_critic_amd.algo.sac import SAC, _freeze
from torch_actor_critic_amd.buffer.visual import VisualReplayBuffer
from torch_actor_critic_amd.envs.visual import MultiObservation
from torch_actor_critic_amd.models.visual import VisualActor, VisualDoubleCritic
from torch_actor_critic_amd.optim import FlatAdam
from torch_actor_critic_amd.parallel.flat import flatten_module_like
from torch_actor_critic_amd.ops import functional as Fo

torch.manual_seed(5)
dev = torch.device("cuda:0")
Fo.set_compute_dtype("bf16")
actor = VisualActor(17, 6, (3, 84, 84), [256, 256], act_limit=1.0).to(dev)
critic = VisualDoubleCritic(17, 6, (3, 84, 84), [256, 256]).to(dev)
target = deepcopy(critic); _freeze(target, True)
pi_opt, q_opt = FlatAdam(actor), FlatAdam(critic)
tf = flatten_module_like(target)
buf = VisualReplayBuffer(5000, act_dim=6, device=dev)
mo = MultiObservation(torch.randn(17, device=dev), torch.randn(3,84,84, device=dev))
buf.store(mo, np.zeros(6), 0.0, mo, 0.0)
n = 2000
buf.features[:n].normal_(); buf.next_features[:n].normal_()
buf.frames[:n].random_(0,255); buf.next_frames[:n].random_(0,255)
buf.actions[:n].uniform_(-1,1); buf.rewards[:n].normal_()
buf.size = n; buf.ptr = n % buf.max_size; buf._size_dev.fill_(n)
sac = SAC(alpha=0.2, gamma=0.99, polyak=0.995, reward_scale=1.0, epochs=1,
          batch_size=64, start_steps=0, steps_per_epoch=1, max_ep_len=10,
          update_after=0, update_every=1, save_every=10)

# profile the EAGER phases (what the capture records), op-attributed
g = GraphedSACUpdate.__new__(GraphedSACUpdate)
g.sac, g.actor, g.critic, g.target_critic, g.buffer = sac, actor, critic, target, buf
g.pi_opt, g.q_opt, g.target_flat, g.device, g.world = pi_opt, q_opt, tf, dev, 1
g.batch = buf.make_static_batch(64)
g.loss_q_acc = torch.zeros((), device=dev); g.loss_pi_acc = torch.zeros((), device=dev)
g._wt_cache = {}
g._critic_weights = [p for p in critic.parameters() if p.ndim in (2,4)]
g._actor_weights = [p for p in actor.parameters() if p.ndim in (2,4)]
Fo.set_graph_opt(g._wt_cache, True)
for _ in range(3):
    g._phase_critic(); g._phase_policy(); g._phase_finish()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    for _ in range(5):
        g._phase_critic(); g._phase_policy(); g._phase_finish()
    torch.cuda.synchronize()
Fo.set_graph_opt(None, False)
print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=28, max_name_column_width=46))
print("\n== aten ops only ==")
for ev in sorted(prof.key_averages(), key=lambda e: -e.self_device_time_total):
    if ev.key.startswith("aten::") and ev.self_device_time_total > 0:
        print(f"{ev.self_device_time_total/1000:8.2f}ms {ev.count:5d}  {ev.key}")
