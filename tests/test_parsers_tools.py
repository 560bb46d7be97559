"""Chat template parsers, tool parsers, tool base, code reward."""

import pytest

from rllm_amd.parser.chat_template_parser import (
    ChatTemplateParser,
    DeepseekQwenChatTemplateParser,
    QwenChatTemplateParser,
)
from rllm_amd.parser.tool_parser import QwenToolParser, R1ToolParser, ToolParser
from rllm_amd.rewards.code_reward import RewardCodeFn, extract_code, run_tests
from rllm_amd.tools.tool_base import MultiTool, Tool, ToolCall, ToolOutput
from rllm_amd.utils.tokenizer import ByteTokenizer


def test_byte_tokenizer_roundtrip():
    tok = ByteTokenizer()
    text = "hello <|im_start|>user\nworld<|im_end|>"
    ids = tok.encode(text)
    assert tok.decode(ids) == text
    assert tok.decode(ids, skip_special_tokens=True) == "hello user\nworld"


def test_qwen_chat_parser_format_and_masks():
    tok = ByteTokenizer()
    p = QwenChatTemplateParser(tok)
    msgs = [{"role": "user", "content": "hi"}]
    text = p.format(msgs)
    assert text == "<|im_start|>user\nhi<|im_end|>\n<|im_start|>assistant\n"
    ids = p.encode_messages(msgs)
    assert tok.decode(ids) == text

    msgs2 = msgs + [{"role": "assistant", "content": "yo"}]
    ids2, mask = p.tokenize_and_mask(msgs2)
    # assistant block tokens are masked 1
    assert sum(mask) > 0
    assert len(ids2) == len(mask)
    # user tokens masked 0
    user_len = len(tok.encode(p.format_message(msgs[0])))
    assert all(m == 0 for m in mask[:user_len])


def test_cumulative_tokenize_prefix_extension():
    tok = ByteTokenizer()
    p = QwenChatTemplateParser(tok)
    m1 = [{"role": "user", "content": "a"}, {"role": "assistant", "content": "b"}]
    m2 = m1 + [{"role": "user", "content": "c"}, {"role": "assistant", "content": "d"}]
    ids, mask = p.tokenize_and_mask_cumulative([m1, m2])
    ids_flat, mask_flat = p.tokenize_and_mask(m2)
    assert ids == ids_flat  # prefix-extension property: same token stream
    assert mask == mask_flat


def test_parser_dispatch():
    tok = ByteTokenizer()
    assert isinstance(ChatTemplateParser.get_parser(tok, "deepseek-r1"), DeepseekQwenChatTemplateParser)
    assert isinstance(ChatTemplateParser.get_parser(tok, "qwen2.5"), QwenChatTemplateParser)
    assert isinstance(ToolParser.get_parser("r1"), R1ToolParser)
    assert isinstance(ToolParser.get_parser("qwen"), QwenToolParser)


def test_qwen_tool_parser():
    text = 'thinking... <tool_call>\n{"name": "search", "arguments": {"q": "x"}}\n</tool_call> done'
    calls = QwenToolParser().parse(text)
    assert len(calls) == 1
    assert calls[0].name == "search"
    assert calls[0].arguments == {"q": "x"}


def test_r1_tool_parser():
    text = 'Sure:\n```json\n{"name": "calc", "arguments": {"expr": "1+1"}}\n```'
    calls = R1ToolParser().parse(text)
    assert calls[0].name == "calc"


def test_multitool_dispatch():
    class Adder(Tool):
        name = "add"
        parameters = {"type": "object", "properties": {"a": {"type": "number"}, "b": {"type": "number"}}}

        def forward(self, a, b):
            return ToolOutput(name=self.name, output=a + b)

    mt = MultiTool([Adder()])
    out = mt.execute(ToolCall(name="add", arguments={"a": 2, "b": 3}, id="c1"))
    assert out.output == 5 and out.id == "c1"
    out2 = mt.execute(ToolCall(name="nope", arguments={}))
    assert out2.error


def test_code_reward_stdin_stdout():
    code = "x = int(input())\nprint(x * 2)"
    tests = [{"stdin": "3", "stdout": "6"}, {"stdin": "5", "stdout": "10"}]
    out = run_tests(code, tests)
    assert out.is_correct and out.n_passed == 2


def test_code_reward_assert_style_and_extraction():
    resp = "Here's my solution:\n```python\ndef f(x):\n    return x + 1\n```"
    assert "def f" in extract_code(resp)
    fn = RewardCodeFn()
    out = fn(resp, [{"assert": "assert f(2) == 3"}, {"assert": "assert f(0) == 1"}])
    assert out.is_correct
    out2 = fn(resp, [{"assert": "assert f(2) == 5"}])
    assert not out2.is_correct


def test_code_reward_timeout():
    out = run_tests("while True: pass", [{"assert": "assert True"}], timeout=1.5)
    assert not out.is_correct and out.error == "timeout"


def test_qwen_parser_matches_hf_jinja_template():
    """Render-equivalence vs the real Qwen2 Jinja chat template executed by
    transformers' apply_chat_template (offline: a minimal fast tokenizer
    object carries the template; only rendered STRINGS are compared)."""
    try:
        from tokenizers import Tokenizer, models
        from transformers import PreTrainedTokenizerFast
    except ImportError:
        import pytest

        pytest.skip("transformers/tokenizers unavailable")

    QWEN2_TEMPLATE = (
        "{% for message in messages %}"
        "{{'<|im_start|>' + message['role'] + '\n' + message['content'] + '<|im_end|>' + '\n'}}"
        "{% endfor %}"
        "{% if add_generation_prompt %}{{ '<|im_start|>assistant\n' }}{% endif %}"
    )
    hf_tok = PreTrainedTokenizerFast(
        tokenizer_object=Tokenizer(models.WordLevel({"[UNK]": 0}, unk_token="[UNK]")),
        chat_template=QWEN2_TEMPLATE)

    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.utils.tokenizer import ByteTokenizer

    parser = QwenChatTemplateParser(ByteTokenizer())
    for msgs in (
        [{"role": "system", "content": "be brief"},
         {"role": "user", "content": "hi there"}],
        [{"role": "user", "content": "q1"},
         {"role": "assistant", "content": "a1"},
         {"role": "user", "content": "q2"}],
    ):
        ours = parser.format(msgs, add_generation_prompt=True)
        theirs = hf_tok.apply_chat_template(msgs, tokenize=False, add_generation_prompt=True)
        assert ours == theirs, (ours, theirs)
        ours_nogen = parser.format(msgs, add_generation_prompt=False)
        theirs_nogen = hf_tok.apply_chat_template(msgs, tokenize=False, add_generation_prompt=False)
        assert ours_nogen == theirs_nogen


def test_qwen_vl_parser_multimodal_blocks():
    """Qwen-VL parser (reference chat_template_parser.py:578): image parts
    render vision placeholders; extract_image_data pulls the payloads."""
    from rllm_amd.parser.chat_template_parser import ChatTemplateParser, QwenVLChatTemplateParser
    from rllm_amd.utils.tokenizer import ByteTokenizer

    p = QwenVLChatTemplateParser(ByteTokenizer())
    msgs = [{"role": "user", "content": [
        {"type": "text", "text": "what is in "},
        {"type": "image_url", "image_url": {"url": "data:image/png;base64,AAAA"}},
        {"type": "text", "text": " this image?"},
    ]}]
    text = p.format(msgs, add_generation_prompt=True)
    assert "<|vision_start|><|image_pad|><|vision_end|>" in text
    assert "what is in " in text and " this image?" in text
    assert text.endswith("<|im_start|>assistant\n")
    imgs = QwenVLChatTemplateParser.extract_image_data(msgs)
    assert imgs == ["data:image/png;base64,AAAA"]
    # family dispatch
    assert isinstance(ChatTemplateParser.get_parser(ByteTokenizer(), "qwen2.5-vl-7b"),
                      QwenVLChatTemplateParser)
    # plain-text messages unchanged vs base Qwen parser
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser

    plain = [{"role": "user", "content": "hi"}]
    assert p.format(plain) == QwenChatTemplateParser(ByteTokenizer()).format(plain)


def test_r1_tool_parser_special_tokens():
    """DeepSeek-R1 special-token tool-call wire format (reference
    tool_parser.py:47-186) plus the bare-fence fallback."""
    from rllm_amd.parser.tool_parser import R1ToolParser

    p = R1ToolParser()
    text = ("thinking...\n<｜tool▁calls▁begin｜>\n"
            "<｜tool▁call▁begin｜>function<｜tool▁sep｜>get_weather\n"
            "```json\n{\"city\": \"Paris\"}\n```\n"
            "<｜tool▁call▁end｜>\n"
            "<｜tool▁call▁begin｜>function<｜tool▁sep｜>get_time\n"
            "```json\n{\"tz\": \"UTC\"}\n```\n"
            "<｜tool▁call▁end｜>\n<｜tool▁calls▁end｜>")
    calls = p.parse(text)
    assert [c.name for c in calls] == ["get_weather", "get_time"]
    assert calls[0].arguments == {"city": "Paris"}
    # fallback: plain fenced json
    calls = p.parse('```json\n{"name": "f", "arguments": {"x": 1}}\n```')
    assert calls[0].name == "f" and calls[0].arguments == {"x": 1}
    # malformed args survive as _raw
    calls = p.parse("<｜tool▁calls▁begin｜><｜tool▁call▁begin｜>function<｜tool▁sep｜>g\n"
                    "```json\nnot json\n```\n<｜tool▁call▁end｜><｜tool▁calls▁end｜>")
    assert calls[0].name == "g" and "_raw" in calls[0].arguments
    # tool prompts teach the right wire format
    assert "tool▁calls▁begin" in p.get_tool_prompt("{}")
    from rllm_amd.parser.tool_parser import QwenToolParser

    assert "<tool_call>" in QwenToolParser().get_tool_prompt("{}")


def test_verify_equivalence_detects_divergence(caplog):
    """verify_equivalence flags a parser whose rendering differs from the
    tokenizer's own chat template (reference :50)."""
    try:
        from tokenizers import Tokenizer, models
        from transformers import PreTrainedTokenizerFast
    except ImportError:
        import pytest

        pytest.skip("transformers/tokenizers unavailable")

    from rllm_amd.parser.chat_template_parser import (
        LlamaChatTemplateParser,
        QwenChatTemplateParser,
    )

    tpl = ("{% for message in messages %}"
           "{{'<|im_start|>' + message['role'] + '\n' + message['content'] + '<|im_end|>' + '\n'}}"
           "{% endfor %}"
           "{% if add_generation_prompt %}{{ '<|im_start|>assistant\n' }}{% endif %}")
    hf_tok = PreTrainedTokenizerFast(
        tokenizer_object=Tokenizer(models.WordLevel({"[UNK]": 0}, unk_token="[UNK]")),
        chat_template=tpl)

    assert QwenChatTemplateParser(hf_tok).verify_equivalence() is True
    assert LlamaChatTemplateParser(hf_tok).verify_equivalence() is False
    # tokenizers without a template are unverifiable -> True
    from rllm_amd.utils.tokenizer import ByteTokenizer

    assert QwenChatTemplateParser(ByteTokenizer()).verify_equivalence() is True
