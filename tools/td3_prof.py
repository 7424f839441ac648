import sys, os; sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch, torch.nn as nn
from rl_replicas_amd import envs, ops
from rl_replicas_amd.algorithms import TD3
from rl_replicas_amd.evaluator import Evaluator
from rl_replicas_amd.networks import MLP
from rl_replicas_amd.policies import DeterministicPolicy, RandomPolicy
from rl_replicas_amd.q_function import QFunction
from rl_replicas_amd.replay_buffer import ReplayBuffer
from rl_replicas_amd.samplers import VectorSampler
from rl_replicas_amd.utils import set_seed_for_libraries

set_seed_for_libraries(0)
dev = "cuda"
env = envs.make("HalfCheetah-v4")
obs_dim, act_dim = 17, 6
pnet = MLP([obs_dim, 256, 256, act_dim], activation_function=nn.ReLU, output_activation_function=nn.Tanh).to(dev)
policy = DeterministicPolicy(pnet, ops.make_adam(pnet.parameters(), lr=1e-3))
qs = []
for _ in range(2):
    qn = MLP([obs_dim+act_dim, 256, 256, 1], activation_function=nn.ReLU).to(dev)
    qs.append(QFunction(qn, ops.make_adam(qn.parameters(), lr=1e-3)))
venv = envs.VectorEnv("HalfCheetah-v4", num_envs=10)
model = TD3(policy, RandomPolicy(env.action_space), qs[0], qs[1], env,
            VectorSampler(venv, seed=0, is_continuous=True),
            ReplayBuffer(int(1e5), device=dev), Evaluator(seed=1))
import tempfile, time
model._begin_learn(tempfile.mkdtemp())
model.metrics_manager.stdout = False
# fill buffer
exp = model.sampler.sample(2000, model.exploration_policy)
model.replay_buffer.add_experience(exp)
model.current_total_steps = 2000
model.train(model.replay_buffer, 50, 100)  # warm
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    model.train(model.replay_buffer, 50, 100)
torch.cuda.synchronize()
print(f"TD3 train(50 iters): {(time.perf_counter()-t0)/10*1000:.1f} ms")
