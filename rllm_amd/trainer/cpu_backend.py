"""CPUBackend — full-pipeline training with a TinyTorchLM on CPU
(BASELINE.json config 1: "Qwen2.5-0.5B single-turn GSM8K GRPO via tinker
backend on CPU" — the role the tinker backend plays in the reference:
prove the whole pipeline with an in-process engine and no GPU,
SURVEY.md §3.5).

Rollouts go through the REAL gateway (uvicorn thread + local handler) so
trace capture, enrichment, and session plumbing are exercised end to end.
The update is plain fp32 torch: GRPO ratio-clip loss (reference-impl
formulas) + AdamW.
"""

from __future__ import annotations

import torch

from rllm_amd.engine.agentflow_engine import AgentFlowEngine
from rllm_amd.gateway.manager import GatewayManager
from rllm_amd.gateway.models import GatewayConfig
from rllm_amd.gateway.native_adapter import make_torch_lm_local_handler
from rllm_amd.models.torch_lm import TinyTorchLM
from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
from rllm_amd.trainer.backend_protocol import BackendProtocol
from rllm_amd.trainer.batch import PackedRow, rows_from_groups
from rllm_amd.types import Episode, TrajectoryGroup
from rllm_amd.utils.tokenizer import ByteTokenizer


class CPUBackend(BackendProtocol):
    def __init__(self, agent_flow, evaluator=None, model: TinyTorchLM | None = None,
                 lr: float = 1e-4, eps_clip: float = 0.2, kl_beta: float = 0.0,
                 n_parallel_tasks: int = 16, rollout_max_tokens: int = 16,
                 temperature: float = 1.0, seed: int = 0,
                 gateway_config: GatewayConfig | None = None, distill=None):
        self.agent_flow = agent_flow
        self.evaluator = evaluator
        self.model = model or TinyTorchLM(seed=seed)
        self.ref_model = None
        if kl_beta > 0:
            self.ref_model = TinyTorchLM(vocab_size=self.model.vocab_size,
                                         hidden=self.model.hidden, seed=seed)
            self.ref_model.load_state_dict(self.model.state_dict())
        self.lr = lr
        self.eps_clip = eps_clip
        self.kl_beta = kl_beta
        self.n_parallel_tasks = n_parallel_tasks
        self.rollout_max_tokens = rollout_max_tokens
        self.temperature = temperature
        self.seed = seed
        self.optimizer = torch.optim.AdamW(self.model.parameters(), lr=lr)
        self.tokenizer = ByteTokenizer()
        self.parser = QwenChatTemplateParser(self.tokenizer)
        self.gateway: GatewayManager | None = None
        self.flow_engine: AgentFlowEngine | None = None
        self.weight_version_ref = {"v": 0}
        self.gateway_config = gateway_config
        self.distill = distill

    # ------------------------------------------------------------------
    def init_rollout_engine(self):
        if self.flow_engine is not None:  # idempotent
            return self.flow_engine
        handler = make_torch_lm_local_handler(
            self.model, self.parser, eos_token_id=None, seed=self.seed,
            weight_version_ref=self.weight_version_ref)
        gw_cfg = self.gateway_config or GatewayConfig()
        self.gateway = GatewayManager(
            gw_cfg, local_handler=handler,
            parser=self.parser if gw_cfg.cumulative_token_mode else None)
        self.gateway.start()
        self.flow_engine = AgentFlowEngine(
            self.agent_flow, self.gateway, evaluator=self.evaluator,
            n_parallel_tasks=self.n_parallel_tasks,
            default_sampling_params={"max_tokens": self.rollout_max_tokens,
                                     "temperature": self.temperature})
        return self.flow_engine

    async def generate_episodes(self, tasks, uids=None, is_validation=False) -> list[Episode]:
        return await self.flow_engine.execute_tasks(tasks, uids, is_validation=is_validation)

    def transform_to_backend_batch(self, groups: list[TrajectoryGroup]) -> list[PackedRow]:
        return rows_from_groups(groups)

    def shard_backend_batch(self, batch: list[PackedRow], rank: int, world_size: int) -> list[PackedRow]:
        from rllm_amd.trainer.batch import shard_rows_balanced

        return shard_rows_balanced(batch, world_size)[rank]

    def postprocess_episodes(self, episodes):
        if self.distill is None:
            return {}
        from rllm_amd.trainer.distill import TeacherClient, distill_episodes

        if isinstance(self.distill, TeacherClient):
            return distill_episodes(episodes, self.distill)
        d = dict(self.distill)
        return distill_episodes(episodes, d.pop("teacher"), **d)

    def set_max_response_tokens(self, n: int) -> None:
        self.rollout_max_tokens = int(n)
        if self.flow_engine is not None:
            self.flow_engine.default_sampling_params["max_tokens"] = int(n)

    def update_policy(self, rows: list[PackedRow]) -> dict:
        """Fp32 torch GRPO update (token-mean aggregation)."""
        self.optimizer.zero_grad()
        total_tokens = sum(sum(r.response_mask) for r in rows)
        if total_tokens == 0:
            return {"actor/skipped": 1.0}
        loss_sum = torch.zeros(())
        clip_count = 0
        for row in rows:
            toks = [t % self.model.vocab_size for t in row.tokens]
            mask = torch.tensor(row.response_mask, dtype=torch.bool)
            adv = torch.tensor(row.advantages, dtype=torch.float32)[mask]
            old_lp = torch.tensor(row.rollout_logprobs, dtype=torch.float32)[mask]
            # differentiable logprobs of masked tokens
            ids = torch.tensor([toks], dtype=torch.long)
            logits = self.model.forward(ids)[0]
            logp = torch.log_softmax(logits, -1)
            rows_idx = torch.nonzero(mask, as_tuple=True)[0] - 1  # predict t at row t-1
            tgt = torch.tensor(toks)[rows_idx + 1]
            lp = logp[rows_idx, tgt]
            ratio = torch.exp(lp - old_lp)
            s1 = ratio * adv
            s2 = torch.clamp(ratio, 1 - self.eps_clip, 1 + self.eps_clip) * adv
            pg = -torch.min(s1, s2)
            clip_count += int((s1 > s2).sum())
            if self.kl_beta > 0 and self.ref_model is not None:
                with torch.no_grad():
                    ref_logits = self.ref_model.forward(ids)[0]
                    ref_lp = torch.log_softmax(ref_logits, -1)[rows_idx, tgt]
                d = ref_lp - lp
                pg = pg + self.kl_beta * (torch.exp(d) - d - 1)
            loss_sum = loss_sum + pg.sum()
        loss = loss_sum / total_tokens
        loss.backward()
        gnorm = torch.nn.utils.clip_grad_norm_(self.model.parameters(), 1.0)
        self.optimizer.step()
        return {
            "actor/loss": float(loss.detach()),
            "actor/grad_norm": float(gnorm),
            "actor/clipfrac": clip_count / total_tokens,
            "actor/n_response_tokens": total_tokens,
        }

    def on_policy_updated(self, weight_version: int) -> None:
        self.weight_version_ref["v"] = weight_version

    # ---- checkpointing ----
    def save_checkpoint(self, path: str, step: int) -> None:
        torch.save({"model": self.model.state_dict(),
                    "optimizer": self.optimizer.state_dict()}, f"{path}/actor.pt")

    def load_checkpoint(self, path: str) -> int:
        sd = torch.load(f"{path}/actor.pt", weights_only=True)
        self.model.load_state_dict(sd["model"])
        self.optimizer.load_state_dict(sd["optimizer"])
        return 0

    def shutdown(self) -> None:
        if self.gateway is not None:
            self.gateway.stop()

    def on_train_end(self) -> None:
        self.shutdown()
