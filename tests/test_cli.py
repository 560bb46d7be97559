"""CLI tests via Click runner (reference keeps one per subcommand)."""

import json

from click.testing import CliRunner

from rllm_amd.cli.main import cli


def test_cli_help_lists_commands():
    r = CliRunner().invoke(cli, ["--help"])
    assert r.exit_code == 0
    for cmd in ("train", "eval", "dataset", "view", "bench", "build-kernels"):
        assert cmd in r.output


def test_dataset_register_list_show(tmp_path):
    rows = [{"question": f"q{i}", "answer": str(i)} for i in range(5)]
    src = tmp_path / "rows.json"
    src.write_text(json.dumps(rows))
    reg = str(tmp_path / "registry")
    runner = CliRunner()
    r = runner.invoke(cli, ["dataset", "register", "mymath", str(src), "--registry-dir", reg])
    assert r.exit_code == 0, r.output
    assert "5 rows" in r.output
    r = runner.invoke(cli, ["dataset", "list", "--registry-dir", reg])
    assert "mymath" in r.output
    r = runner.invoke(cli, ["dataset", "show", "mymath", "--registry-dir", reg])
    assert "5 rows" in r.output


def test_view_episode_dir(tmp_path):
    from rllm_amd.types import Episode, Step, Trajectory
    from rllm_amd.utils.episode_logger import EpisodeLogger
    from rllm_amd.workflows.workflow import TerminationReason

    st = Step(prompt_ids=[1], response_ids=[2, 3], logprobs=[-0.1, -0.2],
              chat_completions=[{"role": "user", "content": "q"}], reward=1.0)
    ep = Episode(id="t:0", trajectories=[Trajectory(name="s", steps=[st], reward=1.0)],
                 termination_reason=TerminationReason.ENV_DONE, is_correct=True)
    out = EpisodeLogger(tmp_path).log_episodes([ep], "train", 0, 0)
    r = CliRunner().invoke(cli, ["view", str(out)])
    assert r.exit_code == 0, r.output
    assert "t:0" in r.output and "correct=True" in r.output


def test_train_cpu_smoke(tmp_path):
    """End-to-end CLI training run on the CPU backend."""
    rows = [{"question": f"{i}+{i}", "id": str(i)} for i in range(4)]
    src = tmp_path / "rows.json"
    src.write_text(json.dumps(rows))
    reg = str(tmp_path / "reg")
    runner = CliRunner()
    runner.invoke(cli, ["dataset", "register", "tiny", str(src), "--registry-dir", reg])
    r = runner.invoke(cli, [
        "train", "tiny", "--agent", "tests.helpers.cli_flows:flow",
        "--evaluator", "tests.helpers.cli_flows:ev",
        "--backend", "cpu", "--batch-size", "2", "--rollout-n", "2",
        "--max-steps", "1", "--registry-dir", reg, "--logger", "console",
    ])
    assert r.exit_code == 0, r.output


def test_serve_help_and_gpu_guard():
    from click.testing import CliRunner

    from rllm_amd.cli.main import cli

    r = CliRunner().invoke(cli, ["serve", "--help"])
    assert r.exit_code == 0 and "OpenAI-compatible" in r.output
    import torch

    if not torch.cuda.is_available():
        r2 = CliRunner().invoke(cli, ["serve"])
        assert r2.exit_code != 0  # clean error, not a crash
