from mdi_llm_amd.prompts import (
    Llama2,
    Llama3,
    PromptStyle,
    TinyLlama,
    get_user_prompt,
    has_prompt_style,
    load_prompt_style,
    model_name_to_prompt_style,
    save_prompt_style,
)


def test_style_resolution():
    assert isinstance(model_name_to_prompt_style("Llama-2-7b-chat-hf"), Llama2)
    assert isinstance(
        model_name_to_prompt_style("Meta-Llama-3-8B-Instruct"), Llama3
    )
    assert isinstance(
        model_name_to_prompt_style("TinyLlama-1.1B-Chat-v1.0"), TinyLlama
    )


def test_llama3_template():
    out = Llama3().apply("hi there")
    assert out.startswith("<|begin_of_text|>")
    assert "hi there" in out
    assert out.endswith("<|end_header_id|>\n\n")


def test_prompt_style_save_load(tmp_path):
    save_prompt_style(Llama3(), tmp_path)
    assert has_prompt_style(tmp_path)
    style = load_prompt_style(tmp_path)
    assert isinstance(style, Llama3)


def test_file_prompt_fanout(tmp_path):
    f = tmp_path / "prompts.txt"
    f.write_text("first paragraph\n\nsecond one\n\n\nthird\n")
    prompts = get_user_prompt(f"FILE:{f}", 5)
    assert prompts == ["first paragraph", "second one", "third",
                       "first paragraph", "second one"]


def test_plain_prompt_replication():
    assert get_user_prompt("hello", 3) == ["hello"] * 3
