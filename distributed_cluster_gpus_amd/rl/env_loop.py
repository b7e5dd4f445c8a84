"""Generic environment training loop for CHSAC-AF.

Capability parity with the reference HybridSACTrainer
(simcore/rl/hybrid_sac_trainer.py — which is dead AND crashes if called: it
passes an `a_f=` kwarg its own Transition does not accept, SURVEY Appendix
A.7).  This version works against any env exposing:
    get_obs_vector() -> np.ndarray
    get_action_masks() -> (mask_dc, mask_g)
    step(action_dict) -> (next_obs, reward, done, info)  # info['costs']: dict
"""
from typing import Dict

from .agent import CHSACAgent
from .replay import ReplayRing


class EnvLoopTrainer:
    def __init__(self, agent: CHSACAgent, buffer: ReplayRing,
                 batch_size: int = 256, warmup: int = 256):
        self.agent = agent
        self.buffer = buffer
        self.batch_size = batch_size
        self.warmup = warmup

    def step_env_and_learn(self, env) -> Dict[str, float]:
        obs = env.get_obs_vector()
        mask_dc, mask_g = env.get_action_masks()
        a = self.agent.select_action(obs, mask_dc, mask_g)
        next_obs, r, done, info = env.step(a)
        self.buffer.add(s=obs, s_next=next_obs, a_dc=a["dc"], a_g=a["g"],
                        r=r, costs=info.get("costs", {}), done=done,
                        mask_dc=mask_dc, mask_g=mask_g)
        if self.buffer.size >= max(self.warmup, self.batch_size):
            return self.agent.train_step(self.buffer.sample(self.batch_size))
        return {}
