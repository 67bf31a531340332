import numpy as np

from distrl_llm_amd.rl.rewards import (correctness_reward, count_xml,
                                       extract_xml_answer, reward_function,
                                       soft_format_reward)

GOOD = "<think>\nsome reasoning\n</think>\n<answer>\n42\n</answer>"
GOOD_TRAIL = GOOD + "\nextra trailing text"
BAD = "no tags at all 42"


def test_extract_xml_answer():
    assert extract_xml_answer(GOOD) == "42"
    assert extract_xml_answer("<answer> 7 </answer>") == "7"
    assert extract_xml_answer("x<answer>a</answer>y<answer>b</answer>") == "b"
    assert extract_xml_answer(BAD) == BAD.strip()


def test_correctness():
    r = correctness_reward([GOOD, BAD], ["42", "42"])
    assert r.tolist() == [1.0, 0.0]


def test_soft_format():
    r = soft_format_reward([GOOD, BAD, "<think>a</think> <answer>b</answer>"])
    assert r.tolist() == [0.1, 0.0, 0.1]
    # anchored like re.match: leading text defeats it
    assert soft_format_reward(["x" + GOOD]).tolist() == [0.0]


def test_count_xml():
    # all four tags exactly once, newline-terminated: full 0.2 credit
    v = count_xml("<think>\na\n</think>\n<answer>\nb\n</answer>\n")
    assert abs(v - 0.2) < 1e-9
    # trailing text after the closing tag is penalized at 0.001/char
    v2 = count_xml("<think>\na\n</think>\n<answer>\nb\n</answer>\nxyz")
    assert v2 < v
    assert count_xml("nothing") == 0.0


def test_reward_function_shape():
    r = reward_function([GOOD, BAD], ["42", "41"])
    assert r.shape == (2, 2)
    assert r[0, 1] == 1.0 and r[1, 1] == 0.0
    assert r[0, 0] > 0.2  # 0.1 soft + xml tag credit (minus trailing penalty)
    assert isinstance(r, np.ndarray)


def test_reward_adversarial_strings():
    """Adversarial completions must never crash the reward stack
    (reference reward_functions.py handles these the same way: split on
    literal tags, regex with DOTALL)."""
    from distrl_llm_amd.rl.rewards import extract_xml_answer, reward_function
    cases = [
        "",                                       # empty completion
        "<answer>",                               # unterminated tag
        "</answer><answer>",                      # reversed tags
        "<answer><answer>42</answer></answer>",   # nested tags
        "<think>é中文</think><answer>∞</answer>",  # unicode
        "x" * 10000,                              # long no-tag blob
        "<answer>42</answer><answer>7</answer>",  # two answers
    ]
    r = reward_function(cases, ["42"] * len(cases))
    assert r.shape == (len(cases), 2)
    import numpy as np
    assert np.isfinite(r).all()
    # reference split semantics (reward_functions.py:4-7:
    # text.split("<answer>")[-1].split("</answer>")[0]): the LAST
    # <answer> tag wins
    assert extract_xml_answer("<answer><answer>42</answer></answer>") == "42"
    assert extract_xml_answer("<answer>42</answer><answer>7</answer>") == "7"
