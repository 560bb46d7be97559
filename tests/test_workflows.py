"""Workflow ABC + built-ins + UnifiedWorkflowEngine against a scripted
fake rollout engine (CPU; mirrors reference test style)."""

import asyncio

import pytest

from rllm_amd.engine.rollout.model_output import ModelOutput
from rllm_amd.engine.rollout.rollout_engine import RolloutEngine
from rllm_amd.engine.unified_workflow_engine import UnifiedWorkflowEngine
from rllm_amd.environments.frozenlake import FrozenLakeAgent, FrozenLakeEnv
from rllm_amd.eval.runner import pass_at_k
from rllm_amd.workflows.builtin import MultiTurnWorkflow, SimpleWorkflow
from rllm_amd.workflows.store import InMemoryStore
from rllm_amd.workflows.workflow import TerminationReason


class ScriptedEngine(RolloutEngine):
    """Returns scripted responses in order (cycling)."""

    def __init__(self, responses):
        super().__init__()
        self.responses = responses
        self.i = 0

    async def _get_model_response(self, messages, **kwargs):
        text = self.responses[self.i % len(self.responses)]
        self.i += 1
        return ModelOutput(text=text, content=text,
                           prompt_ids=[1, 2, 3], completion_ids=[9, 9], logprobs=[-0.1, -0.1],
                           finish_reason="stop")


def test_simple_workflow_reward():
    eng = ScriptedEngine(["the answer is 4"])

    def reward_fn(task, response):
        return 1.0 if "4" in response else 0.0

    wf = SimpleWorkflow(eng, reward_fn=reward_fn)
    ep = asyncio.run(wf.run_with_termination_handling({"question": "2+2?"}, "t:0"))
    assert ep.id == "t:0"
    assert ep.is_correct
    assert ep.trajectories[0].reward == 1.0
    assert ep.termination_reason == TerminationReason.ENV_DONE
    assert ep.metrics.get("solver_acc") == 1.0


def test_multi_turn_frozenlake_reaches_goal():
    # path on default map: down down right right down right -> G
    eng = ScriptedEngine(["down", "down", "right", "right", "down", "right"])
    wf = MultiTurnWorkflow(eng, FrozenLakeAgent, FrozenLakeEnv, max_turns=10)
    ep = asyncio.run(wf.run_with_termination_handling({}, "fl:0"))
    assert ep.termination_reason == TerminationReason.ENV_DONE
    traj = ep.trajectories[0]
    assert traj.reward == 1.0
    assert traj.steps[-1].done
    # cumulative chat structure: later steps extend earlier ones
    assert traj.is_cumulative()


def test_multi_turn_hole_ends_episode():
    eng = ScriptedEngine(["right", "down"])  # (0,1) then (1,1)=H
    wf = MultiTurnWorkflow(eng, FrozenLakeAgent, FrozenLakeEnv, max_turns=10)
    ep = asyncio.run(wf.run_with_termination_handling({}, "fl:1"))
    assert ep.trajectories[0].reward == 0.0
    assert ep.termination_reason == TerminationReason.ENV_DONE


def test_workflow_timeout_produces_episode():
    class SlowEngine(ScriptedEngine):
        async def _get_model_response(self, messages, **kwargs):
            await asyncio.sleep(5)
            return await super()._get_model_response(messages, **kwargs)

    wf = SimpleWorkflow(SlowEngine(["x"]), timeout=0.2)
    ep = asyncio.run(wf.run_with_termination_handling({"question": "q"}, "t:0"))
    assert ep.termination_reason == TerminationReason.TIMEOUT


def test_unified_workflow_engine_pool():
    eng = ScriptedEngine(["down", "down", "right", "right", "down", "right"] * 10)
    uwe = UnifiedWorkflowEngine(
        MultiTurnWorkflow,
        workflow_args={"agent_cls": FrozenLakeAgent, "env_cls": FrozenLakeEnv, "max_turns": 10},
        rollout_engine=eng, n_parallel_tasks=2)
    eps = asyncio.run(uwe.execute_tasks([{}, {}, {}], ["a:0", "a:1", "a:2"]))
    assert len(eps) == 3
    assert all(e.id in ("a:0", "a:1", "a:2") for e in eps)


def test_store():
    async def run():
        store = InMemoryStore()
        await store.set("k", 1)
        assert await store.get("k") == 1
        await store.update("k", lambda v: (v or 0) + 1)
        assert await store.get("k") == 2
        await store.delete("k")
        assert await store.get("k", "gone") == "gone"

    asyncio.run(run())


def test_pass_at_k_estimator():
    assert pass_at_k(10, 10, 1) == 1.0
    assert pass_at_k(10, 0, 5) == 0.0
    assert 0 < pass_at_k(10, 3, 1) < 1
    assert pass_at_k(10, 3, 1) == pytest.approx(0.3)


def test_cumulative_workflow_builds_chain():
    from rllm_amd.workflows.builtin import CumulativeWorkflow

    eng = ScriptedEngine(["go", "stop"])

    def env_fn(task, history):
        n_assistant = sum(1 for m in history if m.get("role") == "assistant")
        done = n_assistant >= 2
        return f"state-{n_assistant}", float(done), done

    wf = CumulativeWorkflow(eng, env_fn, max_turns=5)
    ep = asyncio.run(wf.run_with_termination_handling({"q": 1}, "c:0"))
    traj = ep.trajectories[0]
    assert len(traj.steps) == 2
    assert traj.is_cumulative()
    assert traj.reward == 1.0


def test_distillation_workflow_attaches_advantages():
    from rllm_amd.workflows.builtin import DistillationWorkflow

    student = ScriptedEngine(["answer"])

    class Teacher(ScriptedEngine):
        @property
        def supports_token_in_token_out(self):
            return True

        async def get_token_output_from_token_input(self, token_input, **kw):
            class Out:
                logprobs = [-0.05, -0.05]

            return Out()

    wf = DistillationWorkflow(student, Teacher([]), coef=2.0)
    ep = asyncio.run(wf.run_with_termination_handling({"question": "q"}, "d:0"))
    step = ep.trajectories[0].steps[0]
    # student logprobs -0.1 each, teacher -0.05 -> adv = 2*(0.05) = 0.1
    assert step.advantage == pytest.approx([0.1, 0.1])


def test_tool_calling_flow_e2e():
    """ToolCallingFlow (reference ToolCallingMixin): model emits a
    <tool_call>, the loop executes it and feeds the result back; episode
    assembled from the gateway traces has one step per LLM turn."""
    import asyncio
    import json as _json

    from rllm_amd.engine.agentflow_engine import AgentFlowEngine
    from rllm_amd.gateway.manager import GatewayManager
    from rllm_amd.gateway.models import GatewayConfig
    from rllm_amd.tools.python_tool import CalculatorTool
    from rllm_amd.types import Task
    from rllm_amd.workflows.tool_loop import ToolCallingFlow

    turn_log = []

    async def handler(request):
        body = request["body"]
        msgs = body["messages"]
        turn_log.append(len(msgs))
        if not any(m["role"] == "tool" for m in msgs):
            content = ('I will compute it.\n<tool_call>\n'
                       '{"name": "calculator", "arguments": {"expression": "6*7"}}\n'
                       '</tool_call>')
        else:
            tool_msg = next(m for m in msgs if m["role"] == "tool")
            value = _json.loads(tool_msg["content"])["output"]
            content = f"The answer is {value}."
        n = 4
        return {"id": "x", "object": "chat.completion", "model": "m",
                "choices": [{"index": 0,
                             "message": {"role": "assistant", "content": content},
                             "token_ids": list(range(n)), "prompt_token_ids": [1, 2, 3],
                             "logprobs": {"token_logprobs": [-0.1] * n},
                             "finish_reason": "stop"}],
                "usage": {}}

    gw = GatewayManager(GatewayConfig(), local_handler=handler)
    gw.start()
    try:
        flow = ToolCallingFlow([CalculatorTool()], max_turns=4)
        engine = AgentFlowEngine(flow, gw, evaluator=None, n_parallel_tasks=2)
        eps = asyncio.run(engine.execute_tasks(
            [Task(id="t", instruction="what is 6*7?")], ["t:0"]))
    finally:
        gw.stop()

    assert len(turn_log) == 2  # tool turn + answer turn
    ep = eps[0]
    steps = ep.trajectories[0].steps
    assert len(steps) == 2
    assert "tool_call" in steps[0].model_response
    assert "The answer is 42" in steps[1].model_response
    for st in steps:
        assert st.response_ids and len(st.logprobs) == len(st.response_ids)


def test_tool_loop_with_cumulative_token_mode():
    """Multi-turn tool agent THROUGH cumulative token mode: turn N+1's
    prompt ids must prefix-extend turn N's (the training-mask invariant),
    with the tool result appearing as an observation segment."""
    import asyncio
    import json as _json

    from rllm_amd.engine.agentflow_engine import AgentFlowEngine
    from rllm_amd.gateway.manager import GatewayManager
    from rllm_amd.gateway.models import GatewayConfig
    from rllm_amd.gateway.native_adapter import make_torch_lm_local_handler
    from rllm_amd.models.torch_lm import TinyTorchLM
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.tools.python_tool import CalculatorTool
    from rllm_amd.trainer.batch import rows_from_episodes
    from rllm_amd.types import Task
    from rllm_amd.utils.tokenizer import ByteTokenizer

    parser = QwenChatTemplateParser(ByteTokenizer())
    model = TinyTorchLM(seed=0)
    handler = make_torch_lm_local_handler(model, parser, eos_token_id=None, seed=0)

    # wrap: first turn emits a tool call TEXT, later turns delegate to the model
    turn_count = {"n": 0}
    call_txt = ('<tool_call>\n{"name": "calculator", "arguments": {"expression": "6*7"}}\n'
                "</tool_call>")

    async def scripted(request):
        resp = await handler(request)
        turn_count["n"] += 1
        if turn_count["n"] == 1:
            # keep token ids AND text consistent: re-encode the tool-call text
            toks = parser.tokenizer.encode(call_txt)
            choice = resp["choices"][0]
            choice["message"]["content"] = call_txt
            choice["token_ids"] = toks
            choice["logprobs"] = {"token_logprobs": [-0.1] * len(toks)}
        return resp

    gw = GatewayManager(GatewayConfig(cumulative_token_mode=True), local_handler=scripted,
                        parser=parser)
    gw.start()
    try:
        from rllm_amd.workflows.tool_loop import ToolCallingFlow

        flow = ToolCallingFlow([CalculatorTool()], max_turns=3)
        engine = AgentFlowEngine(flow, gw, evaluator=None, n_parallel_tasks=1)
        eps = asyncio.run(engine.execute_tasks([Task(id="t", instruction="6*7?")], ["t:0"]))
    finally:
        gw.stop()

    ep = eps[0]
    steps = ep.trajectories[0].steps
    assert len(steps) >= 2
    # cumulative invariant: turn 2's prompt ids prefix-extend turn 1's
    full1 = steps[0].prompt_ids + steps[0].response_ids
    assert steps[1].prompt_ids[: len(full1)] == full1
    # and the merged training row masks the tool observation as context
    rows = rows_from_episodes(eps)
    assert len(rows) == 1  # prefix merge collapsed the turns into one row
    row = rows[0]
    assert sum(row.response_mask) == sum(len(s.response_ids) for s in steps)
