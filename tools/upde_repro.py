import os, torch
os.environ["GCBF_AMD_UPDATE_CAPTURE_DEBUG"] = "1"
from gcbf_amd.env import make_env
from gcbf_amd.algo import make_algo
from gcbf_amd.rollout import RolloutEngine
from gcbf_amd.trainer.utils import set_seed
from gcbf_amd.utils.amp import enable_bf16
set_seed(11)
dev = torch.device("cuda")
env = make_env("DubinsCar", 16, dev); env.train()
algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim, env.action_dim, dev, batch_size=512)
enable_bf16(algo)
env.reset()
eng = RolloutEngine(env, algo)
for _ in range(512):
    if eng.step(prob=0.7):
        eng.reload()
algo.update(512)
print("engine:", algo._upd_engine)
