"""Training loop (reference gcbf/trainer/trainer.py:15-141).

Additions over the reference:
* an env-steps/sec throughput meter (the headline benchmark metric);
* optional training resume (optimizer + step state in ``trainer_state.pt``,
  a new file beside the reference-compatible checkpoint layout);
* data-parallel awareness: only rank 0 logs/saves; the step loop itself is
  identical on every rank (per-rank envs and buffers, gradient all-reduce
  happens inside ``algo.update`` via the ``grad_sync`` hook).
"""
from __future__ import annotations

import os
import time
from typing import Tuple

import numpy as np
import torch

from ..algo.base import Algorithm
from ..env.base import MultiAgentEnv
from .summary import SummaryWriter, NullWriter


class Trainer:

    def __init__(self, env: MultiAgentEnv, env_test: MultiAgentEnv,
                 algo: Algorithm, log_dir: str, rank: int = 0,
                 world_size: int = 1):
        self.env = env
        self.env_test = env_test
        self.algo = algo
        self.log_dir = log_dir
        self.rank = rank
        self.world_size = world_size

        if rank == 0:
            os.makedirs(log_dir, exist_ok=True)
            self.model_dir = os.path.join(log_dir, "models")
            os.makedirs(self.model_dir, exist_ok=True)
            self.writer = SummaryWriter(log_dir=os.path.join(log_dir,
                                                             "summary"))
        else:
            self.model_dir = os.path.join(log_dir, "models")
            self.writer = NullWriter()

    def _make_engine(self):
        """hipGraph-captured rollout when supported (GPU + HIP ext + GCBF)."""
        try:
            from ..rollout import RolloutEngine, engine_supported
            if engine_supported(self.env, self.algo):
                if self.env.data is None:
                    self.env.reset()
                return RolloutEngine(self.env, self.algo)
        except Exception as e:
            if self.rank == 0:
                print(f"> rollout capture unavailable ({e}); "
                      f"using the eager loop", flush=True)
        return None

    def train(self, steps: int, eval_interval: int, eval_epi: int,
              start_step: int = 1, capture: bool = True):
        start_time = time.time()
        data = self.env.reset()
        engine = self._make_engine() if capture else None
        if engine is not None and self.rank == 0:
            print("> hipGraph-captured rollout engine active", flush=True)
        last_report = start_time
        steps_since_report = 0

        verbose = None
        for step in range(start_step, steps + 1):
            if engine is not None:
                done = engine.step(prob=1 - (step - 1) / steps)
                if done:
                    engine.reload()
            else:
                if data.u_ref is None:
                    data.update(u_ref=self.env.u_ref(data))
                action = self.algo.step(data, prob=1 - (step - 1) / steps)
                next_data, reward, done, info = self.env.step(action)
                next_data.update(u_ref=self.env.u_ref(next_data))
                self.algo.post_step(data, action, reward, done, next_data)
                data = self.env.reset() if done else next_data

            if self.algo.is_update(step):
                verbose = self.algo.update(step, self.writer)

            steps_since_report += 1
            now = time.time()
            if now - last_report > 30 and self.rank == 0:
                rate = steps_since_report / (now - last_report)
                self.writer.add_scalar("perf/env_steps_per_sec",
                                       rate * self.world_size, step)
                print(f"step {step}/{steps} | "
                      f"{rate * self.world_size:.1f} env-steps/s (whole job)",
                      flush=True)
                last_report, steps_since_report = now, 0

            if step % eval_interval == 0:
                if eval_epi > 0 and self.rank == 0:
                    reward_mean, eval_info = self.eval(step, eval_epi)
                    msg = (f"step: {step}, time: "
                           f"{time.time() - start_time:.0f}s, "
                           f"reward: {reward_mean:.2f}")
                    for key, val in eval_info.items():
                        msg += f", {key}: {val}"
                    print(msg, flush=True)
                if verbose is not None and self.rank == 0:
                    print("step: " + str(step) + "".join(
                        f", {k}: {v:.3f}" for k, v in verbose.items()),
                        flush=True)
                if self.rank == 0:
                    self.algo.save(os.path.join(self.model_dir,
                                                f"step_{step}"))
                    self._save_trainer_state(step)
                self.algo._env = self.env

        if self.rank == 0:
            print(f"> Done in {time.time() - start_time:.0f} seconds",
                  flush=True)

    def _save_trainer_state(self, step: int):
        """Resume state (new vs. reference, separate file)."""
        if not hasattr(self.algo, "extra_state"):
            return
        torch.save({"step": step, "algo": self.algo.extra_state()},
                   os.path.join(self.log_dir, "trainer_state.pt"))

    def load_trainer_state(self) -> int:
        """Returns the step to resume from (1 if no state)."""
        path = os.path.join(self.log_dir, "trainer_state.pt")
        if not os.path.exists(path):
            return 1
        state = torch.load(path, map_location=self.algo.device,
                           weights_only=False)
        if hasattr(self.algo, "load_extra_state"):
            self.algo.load_extra_state(state["algo"])
        return state["step"] + 1

    def eval(self, step: int, eval_epi: int) -> Tuple[float, dict]:
        # reference gcbf/trainer/trainer.py:95-141
        rewards = []
        safe_rate = []
        reach = torch.zeros(self.env_test.num_agents)
        self.algo._env = self.env_test
        for _ in range(eval_epi):
            safe_agent = torch.ones(self.env_test.num_agents).bool()
            data = self.env_test.reset()
            epi_reward = 0.0
            while True:
                data.update(u_ref=self.env_test.u_ref(data))
                action = self.algo.apply(data)
                data, reward, done, info = self.env_test.step(action)
                epi_reward += float(reward.float().mean())
                if "collision" in info:
                    safe_agent[info["collision"].cpu()] = False
                if "reach" in info:
                    reach = info["reach"]
                if done:
                    break
            rewards.append(epi_reward)
            safe_rate.append(safe_agent.sum().item()
                             / self.env_test.num_agents)

        self.writer.add_scalar("test/reward", float(np.mean(rewards)), step)
        self.writer.add_scalar("test/safe_rate", float(np.mean(safe_rate)),
                               step)
        return float(np.mean(rewards)), {
            "safe": round(float(np.mean(safe_rate)), 2),
            "reach": round(float(torch.mean(reach.float())), 2)}
