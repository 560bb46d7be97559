"""FrozenLake environment + agent (BASELINE.json config 4: multi-turn
tool agent; mirrors the reference's frozenlake example env family).

Deterministic (non-slippery) grid: S start, F frozen, H hole, G goal.
Actions: left/down/right/up. Reward 1.0 on reaching G, 0 otherwise;
episode ends on H, G.
"""

from __future__ import annotations

import re
from typing import Any

from rllm_amd.agents.agent import BaseAgent
from rllm_amd.environments.base_env import BaseEnv
from rllm_amd.types import Action, Step, Trajectory

DEFAULT_MAP = ["SFFF", "FHFH", "FFFH", "HFFG"]
ACTIONS = {"left": (0, -1), "down": (1, 0), "right": (0, 1), "up": (-1, 0)}


class FrozenLakeEnv(BaseEnv):
    def __init__(self, desc: list[str] | None = None, max_steps: int = 30):
        self.desc = desc or DEFAULT_MAP
        self.max_steps = max_steps
        self.pos = (0, 0)
        self.steps = 0

    def reset(self, task: dict | None = None, **kwargs):
        if task and task.get("desc"):
            self.desc = task["desc"]
        self.pos = (0, 0)
        self.steps = 0
        return self.render(), {}

    def render(self) -> str:
        rows = []
        for r, row in enumerate(self.desc):
            cells = []
            for c, ch in enumerate(row):
                cells.append("P" if (r, c) == self.pos else ch)
            rows.append("".join(cells))
        return "\n".join(rows)

    def step(self, action: Any):
        self.steps += 1
        act = str(action).strip().lower()
        if act not in ACTIONS:
            return self.render(), 0.0, self.steps >= self.max_steps, {"invalid": True}
        dr, dc = ACTIONS[act]
        r = min(max(self.pos[0] + dr, 0), len(self.desc) - 1)
        c = min(max(self.pos[1] + dc, 0), len(self.desc[0]) - 1)
        self.pos = (r, c)
        cell = self.desc[r][c]
        if cell == "G":
            return self.render(), 1.0, True, {"result": "goal"}
        if cell == "H":
            return self.render(), 0.0, True, {"result": "hole"}
        return self.render(), 0.0, self.steps >= self.max_steps, {}

    @staticmethod
    def from_dict(info: dict) -> "FrozenLakeEnv":
        return FrozenLakeEnv(desc=info.get("desc"))


class FrozenLakeAgent(BaseAgent):
    """Text agent: sees the rendered grid, answers with an action word."""

    SYSTEM = ("You are navigating a frozen lake. P is you, F frozen, H hole, G goal. "
              "Reply with exactly one action: left, down, right, or up.")

    def __init__(self):
        self._trajectory = Trajectory(name="frozenlake")
        self._messages: list[dict] = [{"role": "system", "content": self.SYSTEM}]

    @property
    def chat_completions(self) -> list[dict]:
        return self._messages

    @property
    def trajectory(self) -> Trajectory:
        return self._trajectory

    def update_from_env(self, observation, reward, done, info, **kwargs):
        self._messages.append({"role": "user", "content": f"Grid:\n{observation}\nYour move?"})
        # chat_completions stays empty until update_from_model fills it, so a
        # trailing half-step (env done) gets pruned by postprocess_episode.
        self._trajectory.steps.append(Step(observation=observation))

    def update_from_model(self, response: str, **kwargs) -> Action:
        self._messages.append({"role": "assistant", "content": response})
        step = self._trajectory.steps[-1]
        step.model_response = response
        step.chat_completions = list(self._messages)
        m = re.search(r"\b(left|down|right|up)\b", response.lower())
        act = m.group(1) if m else "noop"
        step.action = Action(action=act)
        return step.action

    def reset(self):
        self._trajectory = Trajectory(name="frozenlake")
        self._messages = [{"role": "system", "content": self.SYSTEM}]
