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


def test_reference_style_mapping_breadth():
    """Every reference family maps to its style (ref prompts.py:325-366)."""
    from mdi_llm_amd import prompts as P

    cases = {
        "stablelm-tuned-alpha-7b": P.StableLMAlpha,
        "stablelm-zephyr-3b": P.StableLMZephyr,
        "stablecode-instruct-alpha-3b": P.StableCode,
        "RedPajama-INCITE-7B-Chat": P.TogetherComputerChat,
        "RedPajama-INCITE-7B-Instruct": P.TogetherComputerInstruct,
        "falcon-7b-instruct": P.Falcon,
        "vicuna-13b-v1.5": P.Vicuna,
        "longchat-7b-16k": P.Vicuna,
        "Llama-2-7b-chat-hf-function-calling-v2": P.Llama2FunctionCalling,
        "Llama-2-7b-chat-hf": P.Llama2,
        "Meta-Llama-3-8B-Instruct": P.Llama3,
        "FreeWilly2": P.FreeWilly2,
        "Platypus2-70B-instruct": P.Platypus,
        "Nous-Hermes-Llama2-13b": P.NousResearch,
        "CodeLlama-13b-Instruct-hf": P.CodeLlama,
        "Mistral-7B-Instruct-v0.3": P.MistralInstruct,
        "phi-1_5": P.Phi1,
        "phi-2": P.Phi2,
        "TinyLlama-1.1B-Chat-v1.0": P.TinyLlama,
        "CodeGemma-7b-it": P.Gemma,
        "gemma-2b-it": P.Gemma,
        "Danube2-1.8b-chat": P.H2Oai,
        "pythia-1.4b": P.Default,
    }
    for name, cls in cases.items():
        got = P.model_name_to_prompt_style(name)
        assert isinstance(got, cls), (name, type(got).__name__)


def test_new_styles_apply_and_roundtrip(tmp_path):
    """Each added style wraps a prompt and survives save/load."""
    from mdi_llm_amd import prompts as P

    for key, cls in P.prompt_styles.items():
        style = cls()
        out = style.apply("hello world")
        assert "hello world" in out or key == "noprompt"
        P.save_prompt_style(style, tmp_path)
        loaded = P.load_prompt_style(tmp_path)
        assert type(loaded) is cls
