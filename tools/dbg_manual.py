import os, sys, torch
sys.path.insert(0, "/root/repo")
from distributed_sac_amd.algo import SACEngine
from tests.test_engine import make_batch, small_cfg
sys.path.insert(0, "/root/repo/tests") if "/root/repo/tests" not in sys.path else None

torch.manual_seed(0)
cfg = small_cfg("mtsac")
e1 = SACEngine(cfg, "cuda:0", precision="bf16")
e2 = SACEngine(cfg, "cuda:0", precision="bf16")
e2.actor.load_state_dict(e1.actor.state_dict())
e2.local_critic.load_state_dict(e1.local_critic.state_dict())
for e in (e1, e2):
    e.hard_copy_targets(); e.refresh_bf16()
batch = {k: v.cuda() for k, v in make_batch(cfg, seed=0).items()}
eps = [torch.randn(cfg.batch_size, cfg.action_dim, device="cuda") for _ in range(2)]

# monkeypatch to stop right after critic step and dump state
import distributed_sac_amd.algo.sac as sacmod

os.environ["DSAC_NO_MANUAL"] = "1"
e1._eps_queue = [e.clone() for e in eps]
# run autograd path but capture grads before step: hook adam
g1 = {}
orig_step = type(e1.critic_optimizer).step
def spy_step(self):
    g1.setdefault('grad', self.group.flat_grad.clone())
    return orig_step(self)
type(e1.critic_optimizer).step = spy_step
m1 = e1.update({k: v.clone() for k, v in batch.items()})
type(e1.critic_optimizer).step = orig_step

os.environ["DSAC_NO_MANUAL"] = "0"
e2._eps_queue = [e.clone() for e in eps]
g2 = {}
def spy_step2(self):
    g2.setdefault('grad', self.group.flat_grad.clone())
    return orig_step(self)
type(e2.critic_optimizer).step = spy_step2
m2 = e2.update({k: v.clone() for k, v in batch.items()})
type(e2.critic_optimizer).step = orig_step

ga, gm = g1['grad'], g2['grad']
print("critic grad max diff:", (ga - gm).abs().max().item())
print("critic grad rel:", ((ga - gm).abs().max() / ga.abs().max()).item())
print("critic data max diff:", (e1.critic_group.flat_data - e2.critic_group.flat_data).abs().max().item())
print("actor data max diff:", (e1.actor_group.flat_data - e2.actor_group.flat_data).abs().max().item())
print("losses:", m1, m2)
