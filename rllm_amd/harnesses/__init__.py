from rllm_amd.harnesses import coding_agents as _coding_agents  # registers harnesses
from rllm_amd.harnesses.cli_harness import (
    HARNESS_REGISTRY,
    BaseCliHarness,
    BashHarness,
    CurlChatHarness,
    OracleHarness,
    get_harness,
)
from rllm_amd.harnesses.terminus2 import Terminus2Harness
from rllm_amd.harnesses.coding_agents import (
    AiderHarness,
    ClaudeCodeHarness,
    CodexHarness,
    KimiCliHarness,
    MiniSweAgentHarness,
    OpenCodeHarness,
    QwenCodeHarness,
    ReactHarness,
    ZeroClawHarness,
)

__all__ = [
    "HARNESS_REGISTRY", "BaseCliHarness", "get_harness",
    "BashHarness", "OracleHarness", "CurlChatHarness",
    "MiniSweAgentHarness", "AiderHarness", "ClaudeCodeHarness",
    "CodexHarness", "OpenCodeHarness", "QwenCodeHarness",
    "KimiCliHarness", "ZeroClawHarness", "ReactHarness", "Terminus2Harness",
]
